"""Worker allocation — resolve a scheduled worker's devices into the env /
device-node set its process needs.

Reference: pkg/hypervisor/worker/allocation.go:22-482
(AllocateWorkerDevices: AllocatedDevices → DeviceInfos, partition split,
env composition, device nodes, mounts; RecoverPartitionedWorker; dynamic
isolation validation).
"""
from __future__ import annotations

import os
from dataclasses import dataclass, field
from typing import Dict, List, Optional

from .. import constants as C
from ..allocator.partitioning import cu_mask_for_percent, cu_mask_for_xcds
from .device import DeviceController, DeviceInfo


@dataclass
class WorkerSpec:
    """What the scheduler bound (pod annotations) — the hypervisor's input."""

    namespace: str
    name: str
    gpu_uuids: List[str]
    isolation: str = C.IsolationSoft
    qos: str = C.QosMedium
    tflops_limit: float = 0.0
    vram_limit: int = 0
    compute_percent_limit: float = 0.0
    partition_xcds: List[int] = field(default_factory=list)
    workload: str = ""

    @property
    def key(self) -> str:
        return f"{self.namespace}/{self.name}"


@dataclass
class WorkerAllocation:
    spec: WorkerSpec
    devices: List[DeviceInfo]
    env: Dict[str, str]
    device_nodes: List[str]
    shm_path: str
    up_limit_percent: int


def _default_limiter_lib() -> str:
    return os.path.join(os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))), "_native", "libtfhip_limiter.so")


class AllocationController:
    def __init__(self, devices: DeviceController,
                 device_nodes: Optional[List[str]] = None,
                 shm_root: str = C.ShmRoot,
                 limiter_lib: Optional[str] = None):
        self.devices = devices
        self.limiter_lib = limiter_lib if limiter_lib is not None \
            else _default_limiter_lib()
        self.device_nodes = device_nodes or ["/dev/kfd", "/dev/dri"]
        self.shm_root = shm_root
        self._allocations: Dict[str, WorkerAllocation] = {}

    def allocate(self, spec: WorkerSpec) -> WorkerAllocation:
        devs: List[DeviceInfo] = []
        for u in spec.gpu_uuids:
            d = self.devices.device_by_uuid(u)
            if d is None:
                raise KeyError(f"device {u} not found on this node")
            devs.append(d)
        up_limit = self._compute_up_limit(spec, devs)
        shm_path = os.path.join(self.shm_root, spec.namespace, spec.name, "shm")

        env: Dict[str, str] = {
            C.EnvVisibleDevices: ",".join(str(d.index) for d in devs),
            "ROCR_VISIBLE_DEVICES": ",".join(str(d.index) for d in devs),
            C.EnvPodNamespace: spec.namespace,
            C.EnvPodName: spec.name,
            C.EnvShmPath: shm_path,
            C.EnvIsolationMode: spec.isolation,
        }
        if spec.vram_limit:
            env[C.EnvVramLimit] = str(spec.vram_limit)
        env[C.EnvUpLimitPercent] = str(up_limit)

        if spec.isolation == C.IsolationSoft and self.limiter_lib and \
                os.path.exists(self.limiter_lib):
            # the reference injects the soft limiter via an init container
            # copying libXXX_limiter.so + LD_PRELOAD (compose.go:1576-1612)
            env["LD_PRELOAD"] = self.limiter_lib

        if spec.isolation == C.IsolationHard:
            pct = spec.compute_percent_limit or up_limit
            mask, _ = cu_mask_for_percent(max(pct, 0.5))
            # ROCr syntax "<queue-list>:<cu-list>"; queues 0-15 cover every
            # compute queue the workload creates.
            env[C.EnvCuMask] = f"0-15:{mask}"
            env["TF_CU_RANGES"] = mask
        elif spec.isolation == C.IsolationPartitioned and spec.partition_xcds:
            mask = cu_mask_for_xcds(spec.partition_xcds)
            env[C.EnvCuMask] = f"0-15:{mask}"
            env["TF_CU_RANGES"] = mask

        alloc = WorkerAllocation(
            spec=spec, devices=devs, env=env,
            device_nodes=list(self.device_nodes), shm_path=shm_path,
            up_limit_percent=up_limit)
        self._allocations[spec.key] = alloc
        return alloc

    def deallocate(self, key: str) -> Optional[WorkerAllocation]:
        return self._allocations.pop(key, None)

    def get(self, key: str) -> Optional[WorkerAllocation]:
        return self._allocations.get(key)

    def list(self) -> List[WorkerAllocation]:
        return list(self._allocations.values())

    @staticmethod
    def _compute_up_limit(spec: WorkerSpec, devs: List[DeviceInfo]) -> int:
        """tflops limit / device capacity → percent (reference
        controller.go:382 computeUpLimit)."""

        if spec.compute_percent_limit:
            return max(1, min(100, int(round(spec.compute_percent_limit))))
        if spec.tflops_limit and devs:
            cap = devs[0].fp16_tflops or 2500.0
            return max(1, min(100, int(round(100.0 * spec.tflops_limit / cap))))
        return 100
