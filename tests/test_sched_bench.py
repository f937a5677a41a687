"""Scheduler scale benchmark — the envtest-bench analog.

Reference: test/sched/scheduler_bench_test.go:35-77 (1,000 nodes / 4,000
GPUs / 10,000 pods → ~400-500 pods/s on an M4 Pro) and
gpufit_bench_test.go micro-benches. Here: 1,000 fake nodes × 8 MI355X
each through the full framework cycle (PreFilter→…→PostBind).
"""
import time

import pytest

import tensor_fusion_amd.constants as C
from tensor_fusion_amd.allocator.gpuallocator import GpuAllocator
from tensor_fusion_amd.api.store import Store
from tensor_fusion_amd.api.types import GPU, Node, Pod, Resource
from tensor_fusion_amd.quota.quota_store import QuotaStore
from tensor_fusion_amd.scheduler.framework import Scheduler
from tensor_fusion_amd.scheduler.gpuresources import GPUResourcesFit


def mk_world(store, nodes=1000, gpus_per_node=8):
    for n in range(nodes):
        node = Node()
        node.meta.name = f"n{n:04d}"
        store.create(node)
    alloc = GpuAllocator(store=store, quota=QuotaStore(store))
    for n in range(nodes):
        for i in range(gpus_per_node):
            g = GPU()
            g.meta.name = f"n{n:04d}-g{i}"
            g.status.uuid = g.meta.name
            g.status.node = f"n{n:04d}"
            g.status.pool = "pool-a"
            g.status.capacity = Resource(C.MI355X_BF16_TFLOPS,
                                         C.MI355X_VRAM_BYTES, 100.0)
            g.status.available = Resource(C.MI355X_BF16_TFLOPS,
                                          C.MI355X_VRAM_BYTES, 100.0)
            alloc.upsert_gpu_for_testing(g)
    return alloc


def mk_pod(i):
    p = Pod()
    p.meta.name = f"pod-{i:05d}"
    p.meta.namespace = "bench"
    p.scheduler_name = C.SchedulerName
    p.meta.annotations = {
        C.AnnoTflopsRequest: "300",
        C.AnnoVramRequest: str(24 << 30),
        C.AnnoComputePercentRequest: "12",
    }
    return p


@pytest.mark.slow
def test_scheduler_throughput_1k_nodes():
    store = Store()
    alloc = mk_world(store, nodes=1000, gpus_per_node=8)
    fit = GPUResourcesFit(store, alloc)
    sched = Scheduler(store, [fit])
    n_pods = 500
    pods = [mk_pod(i) for i in range(n_pods)]
    for p in pods:
        store.create(p)
    t0 = time.perf_counter()
    ok = 0
    for p in pods:
        r = sched.schedule_pod(p)
        if r.status == "Success":
            ok += 1
    dt = time.perf_counter() - t0
    rate = ok / dt
    print(f"\nscheduler: {ok}/{n_pods} pods in {dt:.2f}s = {rate:.0f} pods/s "
          f"(1k nodes / 8k GPUs)")
    assert ok == n_pods
    # floor well below the reference's 400-500/s but catches regressions
    assert rate > 50, f"only {rate:.0f} pods/s"


def test_prefilter_latency_micro():
    """Reference gpufit_bench_test.go: PreFilter ~480 µs at scale."""

    store = Store()
    alloc = mk_world(store, nodes=100, gpus_per_node=8)
    fit = GPUResourcesFit(store, alloc)
    from tensor_fusion_amd.scheduler.framework import CycleState
    pod = mk_pod(0)
    # warmup
    for _ in range(3):
        fit.pre_filter(CycleState(), pod)
    n = 20
    t0 = time.perf_counter()
    for _ in range(n):
        fit.pre_filter(CycleState(), pod)
    us = (time.perf_counter() - t0) / n * 1e6
    print(f"\nPreFilter: {us:.0f} µs/op at 100 nodes / 800 GPUs")
    assert us < 100_000


def test_filter_score_latency_micro():
    """Reference gpufit_bench_test.go:15-16: Filter 155 ns/op, Score
    167 ns/op — per-node calls against CycleState. Ours read the
    PreFilter-computed mapping; they must stay sub-microsecond-class."""

    store = Store()
    alloc = mk_world(store, nodes=100, gpus_per_node=8)
    fit = GPUResourcesFit(store, alloc)
    from tensor_fusion_amd.scheduler.framework import CycleState
    pod = mk_pod(0)
    state = CycleState()
    nodes, st = fit.pre_filter(state, pod)
    assert st.ok and nodes
    node = nodes[0]
    n = 20000
    t0 = time.perf_counter()
    for _ in range(n):
        fit.filter(state, pod, node)
    f_ns = (time.perf_counter() - t0) / n * 1e9
    t0 = time.perf_counter()
    for _ in range(n):
        fit.score(state, pod, node)
    s_ns = (time.perf_counter() - t0) / n * 1e9
    print(f"\nFilter: {f_ns:.0f} ns/op, Score: {s_ns:.0f} ns/op "
          f"(reference: 155 / 167 ns on an M4 Pro)")
    assert f_ns < 20_000 and s_ns < 20_000
