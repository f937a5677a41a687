from .base import ControllerManager, Reconciler, Request
from .core import (ClusterReconciler, ConnectionReconciler, GPUNodeReconciler,
                   NodeClaimReconciler, NodeReconciler, PodReconciler,
                   PoolReconciler, WorkloadReconciler, default_controllers,
                   generate_worker_pod, select_worker)
from .defrag import DefragController, DefragPlan

__all__ = ["ControllerManager", "Reconciler", "Request", "default_controllers",
           "ClusterReconciler", "PoolReconciler", "NodeReconciler",
           "GPUNodeReconciler", "WorkloadReconciler", "ConnectionReconciler",
           "PodReconciler", "NodeClaimReconciler", "generate_worker_pod",
           "select_worker", "DefragController", "DefragPlan"]
