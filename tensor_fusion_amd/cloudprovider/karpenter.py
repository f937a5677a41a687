"""Karpenter node provider — provisioning through NodeClaim CRs.

Reference: internal/cloudprovider/karpenter/nodeclaim.go (creates
karpenter.sh/v1 NodeClaims and tracks their status conditions). Rides
the stack's own K8sClient; Karpenter's controller does the actual cloud
work, so this provider is pure CR choreography — create a NodeClaim
with the GPU requirements, watch for Launched/Registered conditions,
delete it to deprovision.
"""
from __future__ import annotations

from typing import Optional

from ..k8s.client import ApiError, K8sClient
from .provider import GPUNodeProvider

GROUP = "karpenter.sh"
VERSION = "v1"


class KarpenterProvider(GPUNodeProvider):
    def __init__(self, client: K8sClient, node_class: str = "mi355x",
                 node_pool: str = "tensor-fusion"):
        self.client = client
        self.node_class = node_class
        self.node_pool = node_pool
        # route karpenter.sh kinds through the generic CRD pathing
        from ..k8s import serde
        serde.PLURALS.setdefault("NodeClaim", "nodeclaims")
        serde.CLUSTER_SCOPED.add("NodeClaim")

    def _wire(self, name: str, claim) -> dict:
        itype = getattr(claim, "instance_type", "mi355x.8g")
        gpus = getattr(claim, "gpu_count", 8) or 8
        return {
            "apiVersion": f"{GROUP}/{VERSION}",
            "kind": "NodeClaim",
            "metadata": {
                "name": name,
                "labels": {"tensor-fusion.ai/managed-by": "tensor-fusion"},
            },
            "spec": {
                "nodeClassRef": {"group": GROUP, "kind": "EC2NodeClass",
                                 "name": self.node_class},
                "requirements": [
                    {"key": "karpenter.sh/nodepool", "operator": "In",
                     "values": [self.node_pool]},
                    {"key": "node.kubernetes.io/instance-type",
                     "operator": "In", "values": [itype]},
                ],
                "resources": {"requests": {"amd.com/gpu": str(gpus)}},
            },
        }

    # ------------------------------------------------------- interface

    def create_node(self, claim) -> str:
        name = (getattr(claim, "name", "")
                or getattr(getattr(claim, "meta", None), "name", "")
                or "tf-nodeclaim")
        name = f"{name}-nc"
        try:
            self.client.create(self._wire(name, claim))
        except ApiError as e:
            if not e.conflict:
                raise
        return name

    def terminate_node(self, instance_id: str) -> None:
        try:
            self.client.delete("NodeClaim", instance_id)
        except ApiError as e:
            if not e.not_found:
                raise

    def node_status(self, instance_id: str) -> Optional[str]:
        nc = self.client.try_get("NodeClaim", instance_id)
        if nc is None:
            return None
        status = nc.get("status", {})
        node = status.get("nodeName", "")
        conds = {c.get("type"): c.get("status")
                 for c in status.get("conditions", [])}
        if node and conds.get("Registered") == "True":
            return node
        return None
