"""Full-stack single-process e2e: webhook → workload → scheduler →
allocator → connection (reference flow §3.1/§3.2, on the embedded store
with fake MI355X GPUs — the envtest analog)."""
import time

import pytest

import tensor_fusion_amd.constants as C
from tensor_fusion_amd.api.types import GPU, Container, GPUPool, Node, Pod, Resource
from tensor_fusion_amd.operator import build_operator


def mk_gpu(name, node, pool="pool-a"):
    g = GPU()
    g.meta.name = name
    g.status.uuid = f"uuid-{name}"
    g.status.node = node
    g.status.pool = pool
    g.status.capacity = Resource(C.MI355X_BF16_TFLOPS, C.MI355X_VRAM_BYTES,
                                 100.0)
    g.status.available = Resource(C.MI355X_BF16_TFLOPS, C.MI355X_VRAM_BYTES,
                                  100.0)
    return g


def mk_world(op, nodes=2, gpus_per_node=8):
    pool = GPUPool()
    pool.meta.name = "pool-a"
    op.store.create(pool)
    for n in range(nodes):
        node = Node()
        node.meta.name = f"node-{n}"
        op.store.create(node)
        for i in range(gpus_per_node):
            op.store.create(mk_gpu(f"node-{n}-g{i}", f"node-{n}"))


def client_pod(name="app-1", annotations=None):
    p = Pod()
    p.meta.name = name
    p.meta.namespace = "default"
    p.meta.labels[C.LabelEnabled] = "true"
    p.meta.annotations.update(annotations or {})
    p.containers = [Container(name="main")]
    return p


class TestE2E:
    def test_remote_pod_full_flow(self):
        op = build_operator()
        mk_world(op)
        pod = client_pod(annotations={
            C.AnnoTflopsRequest: "600",
            C.AnnoVramRequest: str(48 << 30),
        })
        op.admit(pod)
        for _ in range(6):
            op.tick()
        # worker pod created and scheduled onto a GPU node
        workers = [p for p in op.store.list("Pod", namespace="default")
                   if p.meta.labels.get(C.LabelComponent) == C.ComponentWorker]
        assert len(workers) == 1
        w = workers[0]
        assert w.status.node.startswith("node-")
        assert C.AnnoGpuIds in w.meta.annotations
        # allocator accounted for it
        alloc = op.allocator.allocation(w.meta.key)
        assert alloc is not None and alloc.bound
        # connection got a URL pointing at the worker
        conn = op.store.get("TensorFusionConnection", "app-1-conn", "default")
        assert conn.status.connection_url.startswith("native+")
        # pool status reflects the allocation
        pool = op.store.get("GPUPool", "pool-a")
        assert pool.status.total.tflops - pool.status.available.tflops >= 600
        # metrics + TSDB populated
        pts = op.tsdb.query("tf_pool_metrics", "allocated_tflops")
        assert pts and pts[-1][1] >= 600

        # teardown: scaling the workload to 0 deletes workers and
        # releases the devices
        def _scale0(obj):
            obj.replicas = 0
        op.store.patch("TensorFusionWorkload", "app-1-wl", "default", _scale0)
        for _ in range(3):
            op.tick()
        workers = [p for p in op.store.list("Pod", namespace="default")
                   if p.meta.labels.get(C.LabelComponent) == C.ComponentWorker]
        assert workers == []
        assert op.allocator.allocation(w.meta.key) is None

    def test_gang_workload_schedules_atomically(self):
        op = build_operator()
        mk_world(op, nodes=1, gpus_per_node=8)
        pod = client_pod(name="trainer", annotations={
            C.AnnoComputePercentRequest: "100",
            C.AnnoVramRequest: str(200 << 30),
            C.AnnoGpuCount: "1",
            C.AnnoGangEnabled: "true",
            C.AnnoGangMinMembers: "4",
            C.AnnoGangTimeout: "10",
            C.AnnoGangGroupKey: "tp4",
        })
        pod.meta.annotations[C.AnnoGpuCount] = "1"
        op.admit(pod)
        # gang workload: webhook sets replicas = gpu_count when gang enabled;
        # force 4 workers via the workload
        def _r(obj):
            obj.replicas = 4
            obj.profile.gang.min_members = 4
        op.store.patch("TensorFusionWorkload", "trainer-wl", "default", _r)
        deadline = time.time() + 30
        scheduled = []
        while time.time() < deadline:
            op.tick()
            scheduled = [p for p in op.store.list("Pod", namespace="default")
                         if p.meta.labels.get(C.LabelComponent) ==
                         C.ComponentWorker and p.status.node]
            if len(scheduled) == 4:
                break
        assert len(scheduled) == 4, [p.meta.name for p in scheduled]
        wl = op.store.get("TensorFusionWorkload", "trainer-wl", "default")
        assert wl.status.gang.phase == "Scheduled"
        # all on the same (only) node, distinct GPUs
        gpus = [p.meta.annotations[C.AnnoContainerGpus] for p in scheduled]
        assert len(set(gpus)) == 4

    def test_unschedulable_creates_node_claim(self):
        op = build_operator()
        mk_world(op, nodes=1, gpus_per_node=1)
        # ask for 4 GPUs: impossible on a 1-GPU world → expander claims
        pod = client_pod(name="big", annotations={
            C.AnnoComputePercentRequest: "100",
            C.AnnoGpuCount: "4",
            C.AnnoIsLocalGpu: "true",
        })
        op.admit(pod)
        for _ in range(4):
            op.tick()
        claims = op.store.list("GPUNodeClaim")
        assert claims, "expander should have created a node claim"
        # provider eventually binds the claim and a Node appears
        deadline = time.time() + 10
        while time.time() < deadline:
            op.tick()
            claims = op.store.list("GPUNodeClaim")
            if claims and claims[0].status.phase == "Bound":
                break
            time.sleep(0.05)
        assert claims[0].status.phase == "Bound"

    def test_preemption_evicts_lower_qos(self):
        op = build_operator()
        mk_world(op, nodes=1, gpus_per_node=1)
        low = client_pod(name="low", annotations={
            C.AnnoComputePercentRequest: "90",
            C.AnnoVramRequest: str(250 << 30),
            C.AnnoQos: C.QosLow,
            C.AnnoIsLocalGpu: "true",
        })
        op.admit(low)
        for _ in range(4):
            op.tick()
        low_pod = op.store.get("Pod", "low", "default")
        assert low_pod.status.node, "low-QoS pod should schedule first"

        high = client_pod(name="high", annotations={
            C.AnnoComputePercentRequest: "90",
            C.AnnoVramRequest: str(250 << 30),
            C.AnnoQos: C.QosCritical,
            C.AnnoIsLocalGpu: "true",
        })
        op.admit(high)
        deadline = time.time() + 15
        while time.time() < deadline:
            op.tick()
            hp = op.store.get("Pod", "high", "default")
            if hp.status.node:
                break
            time.sleep(0.05)
        hp = op.store.get("Pod", "high", "default")
        assert hp.status.node, "critical pod should preempt"
        lp = op.store.get("Pod", "low", "default")
        assert lp.status.phase == "Failed" or not lp.status.node


class TestRestartRecovery:
    def test_operator_restart_rebuilds_allocations(self, tmp_path):
        """a fresh operator over the persisted store rebuilds
        committed allocations from worker pod annotations."""

        persist = str(tmp_path / "state")
        op = build_operator(persist_dir=persist)
        mk_world(op, nodes=1, gpus_per_node=2)
        pod = client_pod(annotations={
            C.AnnoTflopsRequest: "600",
            C.AnnoVramRequest: str(48 << 30)})
        op.admit(pod)
        for _ in range(6):
            op.tick()
        workers = [p for p in op.store.list("Pod", namespace="default")
                   if p.meta.labels.get(C.LabelComponent) == C.ComponentWorker]
        assert workers and workers[0].status.node
        wkey = workers[0].meta.key
        gpu_used = workers[0].meta.annotations[C.AnnoContainerGpus]

        # "restart": fresh operator over the same persisted store
        op2 = build_operator(persist_dir=persist)
        assert op2.allocator.allocation(wkey) is None
        n = op2.recover()
        assert n == 1
        alloc = op2.allocator.allocation(wkey)
        assert alloc is not None and alloc.bound
        g = op2.allocator.gpu(gpu_used)
        assert g.status.available.tflops <= g.status.capacity.tflops - 600
        # a new pod still schedules correctly against recovered state
        pod2 = client_pod(name="app-2", annotations={
            C.AnnoTflopsRequest: "600",
            C.AnnoVramRequest: str(48 << 30)})
        op2.admit(pod2)
        for _ in range(6):
            op2.tick()
        w2 = [p for p in op2.store.list("Pod", namespace="default")
              if p.meta.labels.get(C.LabelComponent) == C.ComponentWorker
              and p.meta.labels.get(C.LabelWorkload) == "app-2-wl"]
        assert w2 and w2[0].status.node


def test_workload_deletion_releases_capacity():
    """Deleting a TensorFusionWorkload cascades to its worker pods
    (owner GC) and the allocator returns their capacity (§3.5 dealloc
    path, end to end)."""

    op = build_operator()
    mk_world(op)
    pod = client_pod(annotations={C.AnnoTflopsRequest: "100",
                                  C.AnnoVramRequest: str(16 << 30)})
    op.admit(pod)
    for _ in range(6):
        op.tick()
    workers = [p for p in op.store.list("Pod", namespace="default")
               if p.meta.labels.get(C.LabelComponent) == C.ComponentWorker]
    assert workers
    before = sum(g.status.available.vram for g in op.allocator.gpus())
    wl_name = workers[0].meta.labels[C.LabelWorkload]
    op.store.delete("TensorFusionWorkload", wl_name, "default")
    op.tick()
    left = [p for p in op.store.list("Pod", namespace="default")
            if p.meta.labels.get(C.LabelComponent) == C.ComponentWorker
            and p.meta.labels.get(C.LabelWorkload) == wl_name]
    assert left == []
    after = sum(g.status.available.vram for g in op.allocator.gpus())
    assert after > before  # the workers' VRAM came back
