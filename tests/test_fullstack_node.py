"""§3.4/§3.2 full-node e2e on one store: a hypervisor (mock amdsmi
devices) publishes GPU CRs; the operator's controllers/scheduler build on
them; a scheduled worker pod flows back into the hypervisor's worker
controller which creates its shm and computes limits — the complete
node-onboarding + scheduling loop, GPU-less."""
import os

import pytest

import tensor_fusion_amd.constants as C
from tensor_fusion_amd.api.types import Container, GPUPool, Node, Pod
from tensor_fusion_amd.hypervisor.main import build_hypervisor
from tensor_fusion_amd.operator import build_operator


def test_node_onboarding_and_worker_flow(tmp_path):
    op = build_operator()
    pool = GPUPool()
    pool.meta.name = "pool-a"
    op.store.create(pool)
    node = Node()
    node.meta.name = "node-0"
    op.store.create(node)
    op.tick()

    # hypervisor comes up on node-0 with 4 mock MI355X and publishes GPUs
    devices, workers, erl, backend = build_hypervisor(
        node="node-0", mock_devices=4, shm_root=str(tmp_path / "shm"),
        store=op.store, pool="pool-a")
    backend.publish_devices()
    backend._update_gpunode(devices.devices())
    for _ in range(4):
        op.tick()

    gpus = op.store.list("GPU")
    assert len(gpus) >= 4
    gn = op.store.get("GPUNode", "node-0")
    assert gn.status.gpu_count >= 4
    pool = op.store.get("GPUPool", "pool-a")
    assert pool.status.gpu_count >= 4
    # allocator ingested the published capacity
    assert len(op.allocator.gpus(node="node-0")) >= 4

    # a client pod → workload → worker pod scheduled onto node-0
    pod = Pod()
    pod.meta.name = "app-x"
    pod.meta.namespace = "default"
    pod.meta.labels[C.LabelEnabled] = "true"
    pod.meta.annotations[C.AnnoTflopsRequest] = "400"
    pod.meta.annotations[C.AnnoVramRequest] = str(24 << 30)
    pod.containers = [Container(name="main")]
    op.admit(pod)
    for _ in range(6):
        op.tick()
    worker_pods = [p for p in op.store.list("Pod", namespace="default")
                   if p.meta.labels.get(C.LabelComponent) == C.ComponentWorker]
    assert worker_pods and worker_pods[0].status.node == "node-0"
    wp = worker_pods[0]
    assert C.AnnoGpuIds in wp.meta.annotations

    # hypervisor side: the backend turns the scheduled pod into a worker
    # allocation + shm page with the right limits (§3.2 tail)
    spec = backend.worker_spec_from_pod(wp)
    assert spec is not None
    st = workers.add_worker(spec)
    assert os.path.exists(st.shm.path)
    d = st.shm.device(0)
    assert d.active
    assert d.mem_limit_bytes >= 24 << 30
    # ERL governs it: one tick writes a rate for the up-limit
    erl.attach(st.shm)
    erl.tick(dt=0.5)
    d = st.shm.device(0)
    assert d.erl_refill_rate > 0


def test_external_device_plugin_coexistence(tmp_path):
    """Devices claimed by a non-tensor-fusion pod (plain device plugin)
    are marked used_by=external-device-plugin and skipped by the
    allocator — the reference's kubelet-checkpoint coexistence path."""

    op = build_operator()
    pool = GPUPool()
    pool.meta.name = "pool-a"
    op.store.create(pool)
    node = Node()
    node.meta.name = "node-0"
    op.store.create(node)
    devices, workers, erl, backend = build_hypervisor(
        node="node-0", mock_devices=2, shm_root=str(tmp_path / "shm"),
        store=op.store, pool="pool-a")
    backend.publish_devices()
    backend.start()
    op.tick()

    ext = Pod()
    ext.meta.name = "legacy-dp-pod"
    ext.meta.namespace = "other"
    ext.status.node = "node-0"
    ext.status.phase = "Running"
    ext.containers = [Container(name="m",
                                resources={"amd.com/gpu": "1"},
                                env={C.EnvVisibleDevices: "0"})]
    op.store.create(ext)
    op.tick()
    g0 = op.store.get("GPU", "node-0-gpu-0")
    g1 = op.store.get("GPU", "node-0-gpu-1")
    assert g0.status.used_by == "external-device-plugin"
    assert g1.status.used_by == "tensor-fusion"
    # the allocator must refuse device 0 now: asking for EVERY device on
    # the node cannot be satisfied while one is externally held
    # (the mock accel's device count is process-sticky, so derive it)
    n_dev = len(devices.devices())
    from tensor_fusion_amd.api.types import AllocRequest, Resource
    req = AllocRequest(pod_name="p", namespace="d", gpu_count=n_dev,
                       request=Resource(1, 1 << 30, 1),
                       limit=Resource(1, 1 << 30, 1))
    scores, reasons = op.allocator.check_quota_and_filter(req)
    assert scores == {}, scores
    # external pod exits → device returns to the pool
    op.store.delete("Pod", "legacy-dp-pod", "other")
    op.tick()
    g0 = op.store.get("GPU", "node-0-gpu-0")
    assert g0.status.used_by == "tensor-fusion"
