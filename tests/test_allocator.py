"""GpuAllocator: two-phase lifecycle, filters, strategies, quota,
partitioning, preemption simulation, TTL sweep, dirty sync, reconcile."""
import time

import pytest

from tensor_fusion_amd import constants as C
from tensor_fusion_amd.allocator import partitioning
from tensor_fusion_amd.allocator.gpuallocator import (AllocationError,
                                                      GpuAllocator)
from tensor_fusion_amd.allocator.strategy import make_strategy
from tensor_fusion_amd.api.store import Store
from tensor_fusion_amd.api.types import (GPUResourceQuota,
                                         GPUResourceQuotaSpec, ObjectMeta,
                                         Resource,
                                         default_mi355x_partition_templates)
from tensor_fusion_amd.quota.quota_store import QuotaExceeded

from helpers import make_gpu, make_node_gpus, make_request


def fleet_allocator(nodes=2, per_node=8, **kw):
    a = GpuAllocator(**kw)
    for n in range(nodes):
        for g in make_node_gpus(f"node-{n}", count=per_node):
            a.upsert_gpu_for_testing(g)
    return a


def test_filter_and_score_basic():
    a = fleet_allocator()
    req = make_request(tflops=100, vram=16 << 30)
    scores, reasons = a.check_quota_and_filter(req)
    assert set(scores) == {"node-0", "node-1"}
    # gpu_scores are computed lazily at Reserve (pick_gpus); PreFilter
    # returns node-level scores only
    picked = a.pick_gpus(req, "node-0")
    assert len(picked) == req.gpu_count


def test_assume_commit_dealloc_cycle():
    a = fleet_allocator(nodes=1)
    req = make_request(tflops=500, vram=64 << 30)
    scores, _ = a.check_quota_and_filter(req)
    picks = a.pick_gpus(req, "node-0", scores["node-0"].gpu_scores)
    assert len(picks) == 1
    a.assume(req, picks)
    g = a.gpu(picks[0])
    assert g.status.available.tflops == pytest.approx(C.MI355X_BF16_TFLOPS - 500)
    assert "wl-1" in g.status.running_apps

    a.commit(req.pod_key)
    a.notify_bound(req.pod_key)
    assert a.allocation(req.pod_key).bound

    a.dealloc(req.pod_key)
    g = a.gpu(picks[0])
    assert g.status.available.tflops == pytest.approx(C.MI355X_BF16_TFLOPS)
    assert g.status.running_apps == []


def test_rollback_restores_capacity():
    a = fleet_allocator(nodes=1)
    req = make_request(tflops=500, vram=64 << 30)
    picks = a.pick_gpus(req, "node-0")
    a.assume(req, picks)
    a.rollback(req.pod_key)
    assert a.gpu(picks[0]).status.available.vram == C.MI355X_VRAM_BYTES
    assert a.allocation(req.pod_key) is None


def test_double_assume_rejected():
    a = fleet_allocator(nodes=1)
    req = make_request()
    picks = a.pick_gpus(req, "node-0")
    a.assume(req, picks)
    with pytest.raises(AllocationError):
        a.assume(req, picks)


def test_oversubscribe_denied_on_device():
    a = fleet_allocator(nodes=1, per_node=1)
    big = make_request(pod="p-big", vram=C.MI355X_VRAM_BYTES - (8 << 30))
    a.assume(big, a.pick_gpus(big, "node-0"))
    small = make_request(pod="p-small", vram=16 << 30)
    scores, reasons = a.check_quota_and_filter(small)
    assert scores == {}
    assert any("insufficient" in r for r in reasons.values())


def test_multi_gpu_same_node():
    a = fleet_allocator(nodes=2, per_node=4)
    req = make_request(count=4, tflops=200, vram=32 << 30)
    scores, _ = a.check_quota_and_filter(req)
    assert scores  # both nodes have 4 GPUs
    picks = a.pick_gpus(req, "node-0", scores["node-0"].gpu_scores)
    assert len(picks) == 4
    assert len({p for p in picks}) == 4
    a.assume(req, picks)
    for p in picks:
        assert a.gpu(p).status.available.tflops < C.MI355X_BF16_TFLOPS


def test_same_node_filter_excludes_small_nodes():
    a = GpuAllocator()
    for g in make_node_gpus("big", count=8):
        a.upsert_gpu_for_testing(g)
    for g in make_node_gpus("small", count=2):
        a.upsert_gpu_for_testing(g)
    req = make_request(count=4)
    scores, _ = a.check_quota_and_filter(req)
    assert set(scores) == {"big"}


def test_compact_vs_lowload_strategies():
    # one half-used GPU, one empty GPU
    a_compact = GpuAllocator(strategy=make_strategy("CompactFirst"))
    a_spread = GpuAllocator(strategy=make_strategy("LowLoadFirst"))
    for a in (a_compact, a_spread):
        g1 = make_gpu("g1")
        g1.status.available = Resource(tflops=C.MI355X_BF16_TFLOPS / 2,
                                       vram=C.MI355X_VRAM_BYTES // 2,
                                       compute_percent=50.0)
        a.upsert_gpu_for_testing(g1)
        a.upsert_gpu_for_testing(make_gpu("g2"))
    req = make_request(tflops=100, vram=16 << 30)
    assert a_compact.pick_gpus(req, "node-0")[0] == "g1"
    assert a_spread.pick_gpus(req, "node-0")[0] == "g2"


def test_quota_enforcement():
    store = Store()
    q = GPUResourceQuota(
        meta=ObjectMeta(name="default", namespace="teama"),
        spec=GPUResourceQuotaSpec(
            total=Resource(tflops=1000, vram=100 << 30, compute_percent=1e9),
            single_max=Resource(tflops=600, vram=60 << 30, compute_percent=1e9)))
    store.create(q)
    a = fleet_allocator(store=store)

    too_big = make_request(pod="p1", ns="teama", tflops=700, vram=10 << 30)
    with pytest.raises(QuotaExceeded):
        a.check_quota_and_filter(too_big)

    ok1 = make_request(pod="p2", ns="teama", tflops=600, vram=50 << 30)
    a.check_quota_and_filter(ok1)
    a.assume(ok1, a.pick_gpus(ok1, "node-0"))

    # assumed usage counts against the namespace total
    ok2 = make_request(pod="p3", ns="teama", tflops=600, vram=20 << 30)
    with pytest.raises(QuotaExceeded):
        a.check_quota_and_filter(ok2)

    a.dealloc(ok1.pod_key)
    a.check_quota_and_filter(ok2)  # fits again

    # status sync writes usage back
    a.assume(ok2, a.pick_gpus(ok2, "node-0"))
    a.commit(ok2.pod_key)
    a.quota.sync_dirty()
    got = store.get("GPUResourceQuota", "default", "teama")
    assert got.status.used.tflops == pytest.approx(600)


def test_partitioned_allocation_xcd_slots():
    tpls = default_mi355x_partition_templates()
    a = GpuAllocator(partition_templates=tpls)
    a.upsert_gpu_for_testing(make_gpu("g0"))
    # 2-XCD slab: 25% of the card
    req = make_request(pod="pp1", tflops=500, vram=60 << 30, partitioned=True,
                       isolation_mode=C.IsolationPartitioned)
    scores, _ = a.check_quota_and_filter(req)
    assert scores
    alloc = a.assume(req, a.pick_gpus(req, "node-0"))
    assert alloc.partition is not None
    assert alloc.partition.xcds == [0, 1]
    # second identical partition lands on the next slab
    req2 = make_request(pod="pp2", tflops=500, vram=60 << 30, partitioned=True,
                        isolation_mode=C.IsolationPartitioned)
    alloc2 = a.assume(req2, ["g0"])
    assert alloc2.partition.xcds == [2, 3]
    # device with both partitions released cleanly
    a.dealloc(req.pod_key)
    g = a.gpu("g0")
    assert len(g.status.allocated_partitions) == 1


def test_partition_template_matching_waste_score():
    tpls = default_mi355x_partition_templates()
    req = make_request(tflops=300, vram=30 << 30)
    t = partitioning.match_partition_template(req, tpls)
    assert t.id == "xcd1"  # smallest that fits
    req_big = make_request(tflops=1300, vram=150 << 30)
    t = partitioning.match_partition_template(req_big, tpls)
    assert t.id == "xcd8"  # 4xcd = 1250 tflops < 1300


def test_cu_mask_generation():
    assert partitioning.cu_mask_for_xcds([0, 1]) == "0-63"
    assert partitioning.cu_mask_for_xcds([2]) == "64-95"
    assert partitioning.cu_mask_for_xcds([0, 2]) == "0-31,64-95"
    mask, cus = partitioning.cu_mask_for_percent(25.0)
    assert cus == 64 and mask == "0-63"
    _, cus1 = partitioning.cu_mask_for_percent(0.1)
    assert cus1 == 1  # 1-CU floor (0.39% granularity)


def test_preemption_simulation():
    a = fleet_allocator(nodes=1, per_node=1)
    low = make_request(pod="lowp", qos=C.QosLow,
                       vram=C.MI355X_VRAM_BYTES - (8 << 30))
    a.assume(low, a.pick_gpus(low, "node-0"))
    a.commit(low.pod_key)

    high = make_request(pod="highp", qos=C.QosHigh, vram=64 << 30)
    scores, _ = a.check_quota_and_filter(high)
    assert scores == {}
    got = a.filter_with_preempt(high)
    assert got is not None
    node, victims = got
    assert node == "node-0" and victims == ["default/lowp"]

    # equal or higher QoS is never preempted
    low2 = make_request(pod="low2", qos=C.QosLow, vram=64 << 30)
    assert a.filter_with_preempt(low2) is None


def test_stale_assumed_sweep_gang_aware():
    a = fleet_allocator(nodes=1)
    a.ASSUME_TTL_S = 0.01
    r1 = make_request(pod="stale1", gang_group="")
    r2 = make_request(pod="ganged", gang_group="gg1")
    a.assume(r1, a.pick_gpus(r1, "node-0"))
    a.assume(r2, a.pick_gpus(r2, "node-0"))
    time.sleep(0.02)
    dropped = a.sweep_stale_assumed(gang_active={"gg1"})
    assert dropped == ["default/stale1"]
    assert a.allocation("default/ganged") is not None


def test_adjust_allocation_vertical_scaling():
    a = fleet_allocator(nodes=1, per_node=1)
    req = make_request(tflops=100, vram=10 << 30)
    a.assume(req, a.pick_gpus(req, "node-0"))
    a.commit(req.pod_key)
    a.adjust_allocation(req.pod_key, Resource(tflops=200, vram=20 << 30))
    g = a.gpu(a.allocation(req.pod_key).gpu_names[0])
    assert g.status.available.vram == C.MI355X_VRAM_BYTES - (20 << 30)
    # shrinking also works
    a.adjust_allocation(req.pod_key, Resource(tflops=50, vram=5 << 30))
    g = a.gpu(a.allocation(req.pod_key).gpu_names[0])
    assert g.status.available.vram == C.MI355X_VRAM_BYTES - (5 << 30)


def test_dirty_sync_to_store_and_reconcile():
    store = Store()
    a = GpuAllocator(store=store)
    for g in make_node_gpus("node-0", count=2):
        store.create(g)
    req = make_request(tflops=100, vram=10 << 30)
    a.assume(req, a.pick_gpus(req, "node-0"))
    a.commit(req.pod_key)
    assert a.sync_dirty() == 1
    got = [g for g in store.list("GPU")
           if g.status.available.vram < C.MI355X_VRAM_BYTES]
    assert len(got) == 1

    # restart: fresh allocator rebuilds from records
    a2 = GpuAllocator(store=store)
    a2.reconcile_from_allocations([(req, a.allocation(req.pod_key).gpu_names)])
    assert a2.allocation(req.pod_key).committed
    # rebuild resets then replays: no double subtraction
    g = a2.gpu(a.allocation(req.pod_key).gpu_names[0])
    assert g.status.available.vram == C.MI355X_VRAM_BYTES - (10 << 30)


def test_adjust_allocation_updates_soa_filter_view():
    """Regression: the numpy SoA fast-filter mirror must see vertical
    scaling immediately — a scale-up that fills the only GPU has to make
    the next filter pass come back empty, with no other state change in
    between to invalidate the mirror."""

    from tensor_fusion_amd.api.types import GPU, AllocRequest

    a = GpuAllocator(store=None)
    g = GPU()
    g.meta.name = "solo-g0"
    g.status.uuid = g.meta.name
    g.status.node = "solo"
    g.status.capacity = Resource(2500.0, 1 << 40, 100.0)
    g.status.available = Resource(2500.0, 1 << 40, 100.0)
    a.upsert_gpu_for_testing(g)
    cap = g.status.capacity

    req = AllocRequest(pod_name="soa", namespace="d",
                       request=Resource(10.0, 1 << 30, 1.0),
                       limit=Resource(cap.tflops, cap.vram, 100.0))
    scores, _ = a.check_quota_and_filter(req)  # builds the SoA mirror
    assert scores
    picked = a.pick_gpus(req, "solo")
    a.assume(req, picked)
    a.commit(req.pod_key)

    # scale the pod up to (almost) the whole GPU's VRAM
    a.adjust_allocation(req.pod_key,
                        Resource(10.0, cap.vram - (1 << 20), 1.0))
    # a half-GPU pod must now be rejected — stale mirror would admit it
    req2 = AllocRequest(pod_name="soa2", namespace="d",
                        request=Resource(10.0, cap.vram // 2, 1.0),
                        limit=Resource(cap.tflops, cap.vram, 100.0))
    scores2, _ = a.check_quota_and_filter(req2)
    assert scores2 == {}, scores2


def test_lazy_node_scores_mapping_semantics():
    """LazyNodeScores must behave exactly like {node: NodeScore} for
    every access pattern the cycle uses (getitem/get/contains/keys/
    values/items/iteration) while storing raw floats until read."""

    from tensor_fusion_amd.allocator.gpuallocator import (LazyNodeScores,
                                                          NodeScore)
    m = LazyNodeScores(zip(["n1", "n2", "n3"], [1.5, 2.5, 0.5]))
    assert set(m) == {"n1", "n2", "n3"} and len(m) == 3
    assert "n2" in m and "nx" not in m
    ns = m["n2"]
    assert isinstance(ns, NodeScore)
    assert ns.node == "n2" and ns.score == 2.5 and ns.gpu_scores == {}
    assert m.get("n2") is ns  # materialized once, cached
    assert m.get("nx") is None
    # values()/items() materialize everything (defrag sorts by .score)
    vals = sorted(m.values(), key=lambda s: s.score, reverse=True)
    assert [v.node for v in vals] == ["n2", "n1", "n3"]
    assert all(isinstance(v, NodeScore) for _, v in m.items())
    assert max(m, key=lambda n: m[n].score) == "n2"
