"""Cluster backend — connects the hypervisor to the control plane's object
store (on k8s this is the kubelet pod watch + GPU CR publication).

Reference: pkg/hypervisor/backend/kubernetes/ — node-scoped pod informer
classifying TF workers (pod_cache.go), GPU CR creation from discovered
devices (mutateGPUResourceState kubernetes_backend.go:312 with capacity
from the hardware table and topology tiers), GPUNode status patches.
"""
from __future__ import annotations

import threading
from typing import List, Optional

from ... import constants as C
from ...api.store import NotFound, Store
from ...api.types import (GPU, GPUNode, GPUStatus, ObjectMeta, Pod, Resource)
from ...utils.resource import parse_quantity, parse_tflops
from ..allocation import WorkerSpec
from ..device import DeviceController, DeviceInfo
from ..worker import WorkerController


class StoreBackend:
    def __init__(self, store: Store, node_name: str,
                 devices: DeviceController, workers: WorkerController,
                 pool: str = ""):
        self.store = store
        self.node_name = node_name
        self.devices = devices
        self.workers = workers
        self.pool = pool
        self._mu = threading.Lock()

    # ------------------------------------------------- GPU CR publication

    def gpu_name(self, dev: DeviceInfo) -> str:
        return f"{self.node_name}-gpu-{dev.index}"

    def publish_devices(self):
        devs = self.devices.devices()
        tiers = self.devices.accel.topology(len(devs)) if devs else []
        for d in devs:
            topo = {}
            for o in devs:
                if o.index != d.index and tiers:
                    topo[o.uuid] = tiers[d.index][o.index]
            cap = Resource(tflops=d.fp16_tflops, vram=d.vram_total,
                           compute_percent=100.0)
            name = self.gpu_name(d)
            existing = self.store.try_get("GPU", name)
            if existing is None:
                self.store.create(GPU(
                    meta=ObjectMeta(name=name,
                                    labels={C.LabelNode: self.node_name}),
                    status=GPUStatus(
                        capacity=cap, available=Resource(cap.tflops, cap.vram,
                                                         cap.compute_percent),
                        uuid=d.uuid, index=d.index, numa_node=d.numa_node,
                        node=self.node_name, pool=self.pool, topology=topo)))
            else:
                def _p(obj, cap=cap, topo=topo, d=d):
                    obj.status.capacity = cap
                    obj.status.topology = topo
                    obj.status.uuid = d.uuid
                    obj.status.phase = "Ready"
                self.store.patch("GPU", name, "", _p)
        self._update_gpunode(devs)

    def _update_gpunode(self, devs: List[DeviceInfo]):
        total = Resource(sum(d.fp16_tflops for d in devs),
                         sum(d.vram_total for d in devs),
                         100.0 * len(devs))
        node = self.store.try_get("GPUNode", self.node_name)
        if node is None:
            n = GPUNode(meta=ObjectMeta(name=self.node_name), pool=self.pool)
            n.status.phase = "Running"
            n.status.total = total
            n.status.gpu_count = len(devs)
            n.status.gpus = [self.gpu_name(d) for d in devs]
            n.status.hypervisor_ready = True
            self.store.create(n)
        else:
            def _p(obj):
                obj.status.phase = "Running"
                obj.status.total = total
                obj.status.gpu_count = len(devs)
                obj.status.gpus = [self.gpu_name(d) for d in devs]
                obj.status.hypervisor_ready = True
            self.store.patch("GPUNode", self.node_name, "", _p)

    # ----------------------------------------------------- worker watch

    def start(self):
        self.publish_devices()
        self.store.on_change("Pod", self._on_pod_event)
        for p in self.store.list("Pod"):
            self._on_pod_event("ADDED", p)

    def _is_my_worker(self, pod: Pod) -> bool:
        return (pod.meta.labels.get(C.LabelComponent) == C.ComponentWorker
                and pod.status.node == self.node_name
                and pod.meta.annotations.get(C.AnnoGpuIds, "") != "")

    def _on_pod_event(self, event: str, pod: Pod):
        if pod.kind != "Pod":
            return
        key = pod.meta.key
        if event == "DELETED" or pod.status.phase in ("Failed", "Succeeded"):
            if self.workers.get(key):
                self.workers.remove_worker(key)
            self._sync_external_usage()
            return
        if self._is_external_gpu_pod(pod):
            # coexistence with a plain device plugin (reference kubelet
            # checkpoint detector, external_dp/kubelet_checkpoint.go):
            # devices claimed outside tensor-fusion are marked so the
            # allocator's phase filter skips them (progressive migration)
            self._sync_external_usage()
            return
        if not self._is_my_worker(pod):
            return
        if self.workers.get(key):
            return
        spec = self.worker_spec_from_pod(pod)
        if spec:
            self.workers.add_worker(spec)

    def _is_external_gpu_pod(self, pod: Pod) -> bool:
        if pod.status.node != self.node_name:
            return False
        if pod.meta.labels.get(C.LabelManaged) == "tensor-fusion":
            return False
        for c in pod.containers:
            if any(r in c.resources for r in ("amd.com/gpu",
                                              "nvidia.com/gpu")):
                return True
        return False

    def _sync_external_usage(self):
        """Mark GPUs consumed by non-tensor-fusion device-plugin pods
        (via their gpu-ids annotation or device index env) as
        used_by=external-device-plugin, and release them when freed."""

        external_idx = set()
        for pod in self.store.list("Pod"):
            if not self._is_external_gpu_pod(pod):
                continue
            if pod.status.phase in ("Failed", "Succeeded"):
                continue
            for c in pod.containers:
                vis = c.env.get(C.EnvVisibleDevices)
                if vis:
                    for tok in vis.split(","):
                        try:
                            external_idx.add(int(tok))
                        except ValueError:
                            continue
        for d in self.devices.devices():
            name = self.gpu_name(d)
            want = ("external-device-plugin" if d.index in external_idx
                    else "tensor-fusion")
            try:
                def _p(obj, want=want):
                    obj.status.used_by = want
                self.store.patch("GPU", name, "", _p)
            except NotFound:
                continue

    def worker_spec_from_pod(self, pod: Pod) -> Optional[WorkerSpec]:
        a = pod.meta.annotations
        uuids = [u for u in a.get(C.AnnoGpuIds, "").split(",") if u]
        if not uuids:
            return None
        part_xcds: List[int] = []
        pid = a.get(C.AnnoPartitionId, "")
        if pid:
            # partition_id format: <gpu>-xcd<N>-<start>
            try:
                parts = pid.rsplit("-", 2)
                n, start = int(parts[-2][3:]), int(parts[-1])
                part_xcds = list(range(start, start + n))
            except (ValueError, IndexError):
                part_xcds = []
        return WorkerSpec(
            namespace=pod.meta.namespace or "default",
            name=pod.meta.name,
            gpu_uuids=uuids,
            isolation=a.get(C.AnnoIsolation, C.IsolationSoft),
            qos=a.get(C.AnnoQos, C.QosMedium),
            tflops_limit=parse_tflops(a.get(C.AnnoTflopsLimit, "0") or "0"),
            vram_limit=int(parse_quantity(a.get(C.AnnoVramLimit, "0") or "0")),
            compute_percent_limit=float(
                a.get(C.AnnoComputePercentLimit, "0") or 0),
            partition_xcds=part_xcds,
            workload=pod.meta.labels.get(C.LabelWorkload, ""),
        )
