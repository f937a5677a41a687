"""The operator's reconcilers.

Reference: internal/controller/ (16 reconcilers). The
hierarchy TensorFusionCluster→GPUPool→GPUNode→GPU is preserved; worker
replica scaling, connection URL publication and pod-lifecycle dealloc
follow internal/controller/tensorfusionworkload_controller.go:62-601,
tensorfusionconnection_controller.go:61-375 and pod_controller.go:81-196.
"""
from __future__ import annotations

import time
from typing import List, Optional

from .. import constants as C
from ..api.store import AlreadyExists, NotFound, Store
from ..api.types import (GPU, Container, GPUNode, GPUPool, Pod, Resource,
                         TensorFusionConnection, TensorFusionWorkload,
                         format_connection_url)
from .base import Reconciler, Request

# --------------------------------------------------------------- cluster


class ClusterReconciler(Reconciler):
    """TensorFusionCluster → owns GPUPools (reference cluster controller)."""

    kind = "TensorFusionCluster"
    watches = ["GPUPool"]

    def reconcile(self, req: Request):
        cluster = self.store.get(self.kind, req.name, req.namespace)
        for tmpl in cluster.pools:
            pool = self.store.try_get("GPUPool", tmpl.meta.name)
            if pool is None:
                p = tmpl.deepcopy()
                p.cluster = cluster.meta.name
                p.meta.owner = f"{self.kind}/{req.namespace}/{req.name}" \
                    if req.namespace else f"{self.kind}//{req.name}"
                try:
                    self.store.create(p)
                except AlreadyExists:
                    pass
        pools = [p for p in self.store.list("GPUPool")
                 if p.cluster == cluster.meta.name]

        def _p(obj):
            obj.status.pool_count = len(pools)
            obj.status.phase = "Running" if pools else "Pending"
        self.store.patch(self.kind, req.name, req.namespace, _p)
        return 0.0


class PoolReconciler(Reconciler):
    """Aggregates node capacity into pool status incl. oversell virtuals
    (reference gpupool_controller.go)."""

    kind = "GPUPool"
    watches = ["GPUNode"]

    def map_event(self, kind, event, obj):
        if kind == "GPUNode" and obj.pool:
            return [Request("GPUPool", obj.pool)]
        return []

    def reconcile(self, req: Request):
        pool = self.store.get(self.kind, req.name)
        nodes = [n for n in self.store.list("GPUNode") if n.pool == req.name]
        total = Resource()
        avail = Resource()
        gpu_count = 0
        for n in nodes:
            total = total.add(n.status.total)
            avail = avail.add(n.status.available)
            gpu_count += n.status.gpu_count
        over = pool.capacity.oversubscription
        vt = Resource(
            tflops=total.tflops * over.tflops_oversell_ratio / 100.0,
            vram=int(total.vram * (100 + over.vram_expand_to_host_mem_percent)
                     / 100.0),
            compute_percent=total.compute_percent
            * over.tflops_oversell_ratio / 100.0)

        def _p(obj):
            s = obj.status
            s.node_count = len(nodes)
            s.gpu_count = gpu_count
            s.total, s.available = total, avail
            s.virtual_total = vt
            s.virtual_available = Resource(
                vt.tflops - (total.tflops - avail.tflops),
                vt.vram - (total.vram - avail.vram),
                vt.compute_percent - (total.compute_percent
                                      - avail.compute_percent))
            s.phase = "Running" if nodes else "Pending"
        self.store.patch(self.kind, req.name, "", _p)
        return 0.0


# ----------------------------------------------------------------- nodes


class NodeReconciler(Reconciler):
    """k8s Node → GPUNode CR for nodes matching a pool's selector
    (reference node_controller.go:220 generateGPUNode); node deletion
    tears down the GPUNode and its GPU inventory (node failure path,
    gpunode_controller.go:376)."""

    kind = "Node"

    def __init__(self, store: Store):
        super().__init__(store)
        store.on_change("Node", self._on_event)

    def _on_event(self, event: str, obj):
        if event != "DELETED":
            return
        name = obj.meta.name
        # deleting the GPU CRs evicts them from the allocator through its
        # own informer (gpuallocator _evict on DELETE)
        for g in list(self.store.list("GPU")):
            if g.status.node == name:
                try:
                    self.store.delete("GPU", g.meta.name)
                except Exception:
                    pass
        try:
            self.store.delete("GPUNode", name)
        except Exception:
            pass

    def reconcile(self, req: Request):
        node = self.store.get(self.kind, req.name)
        self._remove_legacy_taint(node)
        pool = self._match_pool(node)
        if pool is None:
            return 0.0
        gn = self.store.try_get("GPUNode", node.meta.name)
        if gn is None:
            gn = GPUNode()
            gn.meta.name = node.meta.name
            gn.meta.owner = f"Node//{node.meta.name}"
            gn.pool = pool.meta.name
            gn.status.node_ip = node.address
            try:
                self.store.create(gn)
            except AlreadyExists:
                pass
        return 0.0

    def _remove_legacy_taint(self, node) -> bool:
        """Drop the deprecated `tensor-fusion.ai/used-by=tensor-fusion:
        NoSchedule` taint from nodes (reference node_controller.go:300
        removeTensorFusionTaint; skipped while the node is deleting)."""

        if node.meta.deletion_ts is not None:
            return False
        taints = getattr(node, "taints", None) or []
        keep = [t for t in taints
                if not (t.get("key") == C.NodeUsedByTaintKey
                        and t.get("value") == C.TensorFusionSystemName
                        and t.get("effect") == "NoSchedule")]
        if len(keep) == len(taints):
            return False

        def _p(obj):
            obj.taints = keep
        self.store.patch(self.kind, node.meta.name, "", _p)
        return True

    def _match_pool(self, node) -> Optional[GPUPool]:
        for pool in self.store.list("GPUPool"):
            sel = pool.node_manager.node_selector
            if not sel or all(node.labels_.get(k) == v for k, v in sel.items()):
                return pool
        return None


class GPUNodeReconciler(Reconciler):
    """Aggregates GPU CRs into GPUNode status and ensures the per-node
    hypervisor pod exists (reference gpunode_controller.go:570)."""

    kind = "GPUNode"
    watches = ["GPU"]

    def map_event(self, kind, event, obj):
        if kind == "GPU" and obj.status.node:
            return [Request("GPUNode", obj.status.node)]
        return []

    PROBE_NS = "tensor-fusion-sys"

    def _ensure_driver_probe(self, node: str):
        """One-shot ROCm driver probe pod, created BEFORE the hypervisor
        (reference gpunode_controller.go:790-863 runs a probe Job and
        gates hypervisor deployment on its success). Returns the probe's
        rocm-version annotation when it has Succeeded, else None."""

        name = f"driver-probe-{node}"
        probe = self.store.try_get("Pod", name, self.PROBE_NS)
        if probe is None:
            pod = Pod()
            pod.meta.name = name
            pod.meta.namespace = self.PROBE_NS
            pod.meta.labels[C.LabelComponent] = "driver-probe"
            pod.meta.labels[C.LabelNode] = node
            pod.meta.owner = f"GPUNode//{node}"
            pod.containers = [Container(
                name="probe",
                command=["python", "-c",
                         "import torch; assert torch.cuda.is_available();"
                         "print(torch.version.hip)"])]
            pod.status.node = node
            try:
                self.store.create(pod)
            except AlreadyExists:
                pass
            return None
        if probe.status.phase == "Succeeded":
            return probe.meta.annotations.get(
                f"{C.Domain}/rocm-version", "unknown")
        return None

    def reconcile(self, req: Request):
        gn = self.store.get(self.kind, req.name)
        gpus = [g for g in self.store.list("GPU")
                if g.status.node == req.name]
        total = Resource()
        avail = Resource()
        for g in gpus:
            total = total.add(g.status.capacity)
            avail = avail.add(g.status.available)

        # driver probe gates hypervisor rollout — unless devices are
        # already discovered (working driver is then self-evident, e.g.
        # operator restart over a live fleet)
        rocm = self._ensure_driver_probe(req.name)
        if rocm is None and not gpus:
            def _pending(obj):
                obj.status.phase = "Pending"
                obj.status.hypervisor_ready = False
            self.store.patch(self.kind, req.name, "", _pending)
            return 2.0  # re-check until the probe succeeds

        hyp_name = f"hypervisor-{req.name}"
        hyp = self.store.try_get("Pod", hyp_name, "tensor-fusion-sys")
        if hyp is None:
            pool = self.store.try_get("GPUPool", gn.pool) if gn.pool                 else None
            pod = Pod()
            pod.meta.name = hyp_name
            pod.meta.namespace = "tensor-fusion-sys"
            pod.meta.labels[C.LabelComponent] = C.ComponentHypervisor
            pod.meta.labels[C.LabelNode] = req.name
            if pool is not None:
                pod.meta.labels[C.LabelPool] = pool.meta.name
            pod.meta.owner = f"GPUNode//{req.name}"
            # provider env + isolation args from the pool's component
            # config (reference gpunode_controller.go:1029-1143)
            iso = (pool.node_manager.isolation_default
                   if pool is not None else C.IsolationSoft)
            env = {
                "TF_NODE_NAME": req.name,
                "TF_ACCELERATOR_LIB": f"/opt/tensor-fusion/"
                                      f"{C.AcceleratorLibName}",
                "TF_ISOLATION_DEFAULT": iso,
                "HSA_ENABLE_IPC_MODE_LEGACY": "0",
            }
            tmpl = (pool.components.hypervisor_template
                    if pool is not None else {}) or {}
            env.update({k: str(v) for k, v in
                        (tmpl.get("env") or {}).items()})
            image = (pool.components.hypervisor_image
                     if pool is not None else "")
            pod.containers = [Container(
                name="hypervisor",
                image=image or "tensor-fusion/hypervisor:latest",
                command=["python", "-m",
                         "tensor_fusion_amd.hypervisor.main",
                         "--node", req.name,
                         "--backend", "kubernetes",
                         "--device-plugin"],
                env=env)]
            pod.status.node = req.name  # host pod, not scheduled by us
            pod.status.phase = "Running"
            try:
                self.store.create(pod)
            except AlreadyExists:
                pass

        def _p(obj):
            s = obj.status
            s.gpu_count = len(gpus)
            s.gpus = sorted(g.meta.name for g in gpus)
            s.total, s.available = total, avail
            s.hypervisor_ready = True
            if rocm and rocm != "unknown":
                s.rocm_version = rocm
            s.phase = "Running" if gpus else "Pending"
        self.store.patch(self.kind, req.name, "", _p)
        return 0.0


# -------------------------------------------------------------- workload


def generate_worker_pod(wl: TensorFusionWorkload, index: int) -> Pod:
    """Reference internal/worker/worker.go:32 GenerateWorkerPod +
    utils/compose.go:1542 AddWorkerConfAfterTemplate."""

    prof = wl.profile
    pod = Pod()
    # dealloc-before-delete guard (reference pod_controller.go:223
    # handleWorkerPodFinalizer): PodReconciler strips it after dealloc
    pod.meta.finalizers = [C.Finalizer]
    pod.meta.name = f"{wl.meta.name}-worker-{index}"
    pod.meta.namespace = wl.meta.namespace
    pod.meta.labels = {
        C.LabelComponent: C.ComponentWorker,
        C.LabelWorkload: wl.meta.name,
        C.LabelEnabled: "true",
    }
    a = pod.meta.annotations
    r = prof.resources
    a[C.AnnoTflopsRequest] = str(r.requests.tflops)
    a[C.AnnoTflopsLimit] = str(r.limits.tflops)
    a[C.AnnoVramRequest] = str(int(r.requests.vram))
    a[C.AnnoVramLimit] = str(int(r.limits.vram))
    if r.requests.compute_percent:
        a[C.AnnoComputePercentRequest] = str(r.requests.compute_percent)
        a[C.AnnoComputePercentLimit] = str(r.limits.compute_percent)
    a[C.AnnoQos] = prof.qos
    a[C.AnnoIsolation] = prof.isolation_mode
    a[C.AnnoGpuCount] = str(prof.gpu_count)
    if prof.gpu_model:
        a[C.AnnoGpuModel] = prof.gpu_model
    if prof.gang.enabled:
        a[C.AnnoGangEnabled] = "true"
        a[C.AnnoGangMinMembers] = str(prof.gang.min_members or wl.replicas)
        a[C.AnnoGangTimeout] = str(prof.gang.timeout_s)
        a[C.AnnoGangGroupKey] = prof.gang.group_key or wl.meta.name
    pod.scheduler_name = C.SchedulerName
    pod.containers = [Container(
        name="worker",
        command=["tf_vgpu_worker"],
        env={
            C.EnvPodNamespace: wl.meta.namespace,
            C.EnvIsolationMode: prof.isolation_mode,
            C.EnvHypervisorPort: str(C.HypervisorHTTPPort),
        })]
    pod.meta.owner = f"TensorFusionWorkload/{wl.meta.namespace}/{wl.meta.name}"
    return pod


class WorkloadReconciler(Reconciler):
    """Worker replica scaling + gang status (reference
    tensorfusionworkload_controller.go:180-338, :468)."""

    kind = "TensorFusionWorkload"
    watches = ["Pod"]

    def reconcile(self, req: Request):
        wl = self.store.get(self.kind, req.name, req.namespace)
        workers = [p for p in self.store.list("Pod", namespace=req.namespace)
                   if p.meta.labels.get(C.LabelWorkload) == req.name
                   and p.meta.labels.get(C.LabelComponent) == C.ComponentWorker
                   and p.meta.deletion_ts is None]
        # scale up
        have = {p.meta.name for p in workers}
        i = 0
        while len(have) < wl.replicas:
            pod = generate_worker_pod(wl, i)
            if pod.meta.name in have:
                i += 1
                continue
            try:
                self.store.create(pod)
                have.add(pod.meta.name)
            except AlreadyExists:
                have.add(pod.meta.name)
            i += 1
            if i > wl.replicas + len(workers) + 64:
                break
        # scale down (highest index first, prefer unscheduled)
        if len(workers) > wl.replicas:
            workers.sort(key=lambda p: (p.status.node != "", p.meta.name))
            for p in workers[:len(workers) - wl.replicas]:
                try:
                    self.store.delete("Pod", p.meta.name, p.meta.namespace)
                except NotFound:
                    pass

        workers = [p for p in self.store.list("Pod", namespace=req.namespace)
                   if p.meta.labels.get(C.LabelWorkload) == req.name
                   and p.meta.labels.get(C.LabelComponent) == C.ComponentWorker]
        ready = [p for p in workers if p.status.phase in ("Scheduled", "Running")]

        def _p(obj):
            s = obj.status
            s.replicas = len(workers)
            s.ready_replicas = len(ready)
            s.worker_pods = sorted(p.meta.name for p in workers)
            s.phase = "Running" if len(ready) >= obj.replicas else "Pending"
            if obj.profile.gang.enabled:
                s.gang.group = obj.profile.gang.group_key or obj.meta.name
                s.gang.members_total = obj.replicas
                s.gang.members_scheduled = len(ready)
                s.gang.phase = ("Scheduled" if len(ready) >= obj.replicas
                                else "Pending")
        self.store.patch(self.kind, req.name, req.namespace, _p)
        return 0.0


# ------------------------------------------------------------ connection


def select_worker(store: Store, workload: TensorFusionWorkload
                  ) -> Optional[Pod]:
    """Least-connections worker selection with max-skew preference
    (reference internal/worker/worker.go:89 SelectWorker)."""

    workers = [p for p in store.list("Pod", namespace=workload.meta.namespace)
               if p.meta.labels.get(C.LabelWorkload) == workload.meta.name
               and p.meta.labels.get(C.LabelComponent) == C.ComponentWorker
               and p.status.phase in ("Scheduled", "Running")]
    if not workers:
        return None
    conns = store.list("TensorFusionConnection",
                       namespace=workload.meta.namespace)
    load = {p.meta.name: 0 for p in workers}
    for c in conns:
        if c.status.worker in load:
            load[c.status.worker] += 1
    workers.sort(key=lambda p: (load[p.meta.name], p.meta.name))
    return workers[0]


class ConnectionReconciler(Reconciler):
    """TensorFusionConnection → pick a worker, publish connection URL
    (reference tensorfusionconnection_controller.go:61-375)."""

    kind = "TensorFusionConnection"
    watches = ["Pod"]

    def map_event(self, kind, event, obj):
        if kind != "Pod" or \
                obj.meta.labels.get(C.LabelComponent) != C.ComponentWorker:
            return []
        wl = obj.meta.labels.get(C.LabelWorkload, "")
        return [Request(self.kind, c.meta.name, c.meta.namespace)
                for c in self.store.list(self.kind)
                if c.workload == wl]

    def reconcile(self, req: Request):
        conn = self.store.get(self.kind, req.name, req.namespace)
        wl = self.store.try_get("TensorFusionWorkload", conn.workload,
                                req.namespace)
        if wl is None:
            return 1.0
        # keep the current worker while it lives (failover only on loss)
        if conn.status.worker:
            cur = self.store.try_get("Pod", conn.status.worker, req.namespace)
            if cur is not None and cur.status.phase in ("Scheduled", "Running"):
                return 0.0
        worker = select_worker(self.store, wl)
        if worker is None:
            return 1.0
        url = format_connection_url(
            worker.status.pod_ip or worker.status.host_ip or "127.0.0.1",
            C.WorkerPort, worker.meta.name, worker.meta.resource_version)

        def _p(obj):
            obj.status.worker = worker.meta.name
            obj.status.connection_url = url
            obj.status.phase = "Ready"
        self.store.patch(self.kind, req.name, req.namespace, _p)
        return 0.0


# ------------------------------------------------------------------ pods


class PodReconciler(Reconciler):
    """Pod lifecycle: dealloc on delete, TensorFusionConnection creation
    for client pods (reference pod_controller.go:81-196, :262)."""

    kind = "Pod"

    def __init__(self, store: Store, allocator=None):
        super().__init__(store)
        self.allocator = allocator
        store.on_change("Pod", self._on_event)

    def _on_event(self, event: str, obj: Pod):
        if event == "DELETED" and self.allocator is not None:
            try:
                self.allocator.dealloc(obj.meta.key)
            except Exception:
                pass

    def reconcile(self, req: Request):
        pod = self.store.get(self.kind, req.name, req.namespace)
        if pod.meta.deletion_ts and C.Finalizer in pod.meta.finalizers:
            # finalizer path: release the allocation BEFORE the object
            # can disappear, then let the apiserver finish the delete
            if self.allocator is not None:
                try:
                    self.allocator.dealloc(pod.meta.key)
                except Exception:
                    pass

            def _strip(obj):
                obj.meta.finalizers = [f for f in obj.meta.finalizers
                                       if f != C.Finalizer]
            self.store.patch(self.kind, req.name, req.namespace, _strip)
            return 0.0
        if pod.meta.labels.get(C.LabelComponent) == C.ComponentClient:
            conn_name = pod.containers[0].env.get(
                C.EnvConnectionName, f"{req.name}-conn") if pod.containers \
                else f"{req.name}-conn"
            if self.store.try_get("TensorFusionConnection", conn_name,
                                  req.namespace) is None:
                conn = TensorFusionConnection()
                conn.meta.name = conn_name
                conn.meta.namespace = req.namespace
                conn.meta.owner = f"Pod/{req.namespace}/{req.name}"
                conn.workload = pod.meta.labels.get(C.LabelWorkload, "")
                conn.client_pod = req.name
                try:
                    self.store.create(conn)
                except AlreadyExists:
                    pass
        return 0.0


# ------------------------------------------------------------ node claims


class NodeClaimReconciler(Reconciler):
    """GPUNodeClaim → cloud provider create; Bound when the node joins
    (reference gpunodeclaim controller + cloudprovider)."""

    kind = "GPUNodeClaim"

    def __init__(self, store: Store, provider=None):
        super().__init__(store)
        self.provider = provider

    def reconcile(self, req: Request):
        claim = self.store.get(self.kind, req.name)
        if claim.status.phase == "Bound":
            return 0.0
        if self.provider is None:
            return 0.0
        if claim.status.phase == "Pending":
            inst = self.provider.create_node(claim)

            def _p(obj):
                obj.status.phase = "Creating"
                obj.status.instance_id = inst
            self.store.patch(self.kind, req.name, "", _p)
            return 0.5
        # Creating: poll provider
        node_name = self.provider.node_status(claim.status.instance_id)
        if node_name:
            def _p2(obj):
                obj.status.phase = "Bound"
                obj.status.node_name = node_name
            self.store.patch(self.kind, req.name, "", _p2)
            return 0.0
        return 0.5


def default_controllers(store: Store, allocator=None, provider=None
                        ) -> List[Reconciler]:
    from .provider import (ProviderConfigReconciler, ProviderManager,
                           SchedulingConfigReconciler)
    return [
        ClusterReconciler(store),
        PoolReconciler(store),
        NodeReconciler(store),
        GPUNodeReconciler(store),
        WorkloadReconciler(store),
        ConnectionReconciler(store),
        PodReconciler(store, allocator=allocator),
        NodeClaimReconciler(store, provider=provider),
        ProviderConfigReconciler(store,
                                 ProviderManager(allocator=allocator)),
        SchedulingConfigReconciler(store, allocator=allocator),
    ]
