"""Property-based allocator invariants (hypothesis): arbitrary
interleavings of the two-phase lifecycle (filter → assume → commit/
rollback → notify_bound → dealloc) must conserve capacity exactly and
never oversubscribe any GPU — the reference defends this with dedicated
race tests and idempotent unique-allocation maps
(gpuallocator.go:300-316); here the whole operation space is fuzzed."""
import pytest
from hypothesis import given, settings
from hypothesis import strategies as st

import tensor_fusion_amd.constants as C
from tensor_fusion_amd.allocator.gpuallocator import (AllocationError,
                                                      GpuAllocator)
from tensor_fusion_amd.api.types import GPU, AllocRequest, Resource

N_GPUS = 8
CAP_T, CAP_V = 2500.0, C.MI355X_VRAM_BYTES


def fleet():
    a = GpuAllocator(store=None)
    for i in range(N_GPUS):
        g = GPU()
        g.meta.name = f"n0-g{i}"
        g.status.uuid = g.meta.name
        g.status.node = "n0"
        g.status.capacity = Resource(CAP_T, CAP_V, 100.0)
        g.status.available = Resource(CAP_T, CAP_V, 100.0)
        a.upsert_gpu_for_testing(g)
    return a


# ops: (kind, pod_id, tflops_frac_steps, vram_frac_steps, gpu_count)
op = st.tuples(
    st.sampled_from(["alloc", "commit", "rollback", "bind", "dealloc"]),
    st.integers(min_value=0, max_value=11),
    st.integers(min_value=1, max_value=10),
    st.integers(min_value=1, max_value=10),
    st.integers(min_value=1, max_value=2),
)


@settings(max_examples=120, deadline=None)
@given(st.lists(op, min_size=1, max_size=60))
def test_lifecycle_interleavings_conserve_capacity(ops):
    a = fleet()
    state = {}  # pod_key -> phase ("assumed" | "committed" | "bound")
    for kind, pid, tf, vf, count in ops:
        key = f"prop/p{pid}"
        if kind == "alloc" and key not in state:
            req = AllocRequest(
                pod_name=f"p{pid}", namespace="prop", gpu_count=count,
                request=Resource(CAP_T * tf / 10, int(CAP_V * vf / 10),
                                 10.0 * tf),
                limit=Resource(CAP_T, CAP_V, 100.0))
            try:
                scores, _ = a.check_quota_and_filter(req)
                if not scores:
                    continue
                node = max(scores, key=lambda n: scores[n].score)
                picked = a.pick_gpus(req, node)
                a.assume(req, picked)
                state[key] = "assumed"
            except (AllocationError, KeyError):
                continue
        elif kind == "commit" and state.get(key) == "assumed":
            a.commit(key)
            state[key] = "committed"
        elif kind == "rollback" and state.get(key) == "assumed":
            a.rollback(key)
            del state[key]
        elif kind == "bind" and state.get(key) == "committed":
            a.notify_bound(key)
            state[key] = "bound"
        elif kind == "dealloc" and state.get(key) in ("committed", "bound"):
            a.dealloc(key)
            del state[key]

        # invariant 1: no GPU is ever oversubscribed or over-returned
        for g in a.gpus():
            assert -1e-6 <= g.status.available.tflops <= CAP_T + 1e-6, \
                g.meta.name
            assert 0 <= g.status.available.vram <= CAP_V, g.meta.name

    # invariant 2: after releasing everything, capacity returns exactly
    for key, phase in list(state.items()):
        if phase == "assumed":
            a.rollback(key)
        else:
            a.dealloc(key)
    for g in a.gpus():
        assert abs(g.status.available.tflops - CAP_T) < 1e-6
        assert g.status.available.vram == CAP_V
        assert abs(g.status.available.compute_percent - 100.0) < 1e-6


@settings(max_examples=60, deadline=None)
@given(st.lists(st.integers(min_value=1, max_value=10), min_size=1,
                max_size=24))
def test_greedy_fill_never_exceeds_device_capacity(fracs):
    """Greedily admit pods of arbitrary VRAM fractions until filters say
    no; total admitted per GPU must fit its capacity."""

    a = fleet()
    admitted = []
    for i, f in enumerate(fracs):
        req = AllocRequest(
            pod_name=f"g{i}", namespace="prop",
            request=Resource(10.0, int(CAP_V * f / 10), 1.0),
            limit=Resource(CAP_T, CAP_V, 100.0))
        scores, _ = a.check_quota_and_filter(req)
        if not scores:
            continue
        node = max(scores, key=lambda n: scores[n].score)
        try:
            picked = a.pick_gpus(req, node)
            a.assume(req, picked)
            a.commit(req.pod_key)
            admitted.append((req.pod_key, picked[0], int(CAP_V * f / 10)))
        except (AllocationError, KeyError):
            continue
    per_gpu = {}
    for _, gpu, vram in admitted:
        per_gpu[gpu] = per_gpu.get(gpu, 0) + vram
    for gpu, total in per_gpu.items():
        assert total <= CAP_V, (gpu, total)


@settings(max_examples=100, deadline=None)
@given(st.lists(
    st.tuples(st.sampled_from(["check_assume", "commit", "forget",
                               "release"]),
              st.integers(min_value=0, max_value=7),
              st.integers(min_value=1, max_value=6)),
    min_size=1, max_size=50))
def test_quota_store_never_exceeds_namespace_total(ops):
    """Fuzzed two-phase quota lifecycle: committed+assumed never exceeds
    the namespace total when every admission goes through check(), and
    full release drains usage to zero (reference quota_store.go:400-456
    assumed-usage overlay semantics)."""

    from tensor_fusion_amd.api.store import Store
    from tensor_fusion_amd.api.types import GPUResourceQuota
    from tensor_fusion_amd.quota.quota_store import (QuotaExceeded,
                                                     QuotaStore)

    store = Store()
    q = GPUResourceQuota()
    q.meta.name = "quota"
    q.meta.namespace = "fuzz"
    q.spec.total = Resource(1000.0, 100 << 30, 0.0)
    q.spec.max_workers = 6
    store.create(q)
    qs = QuotaStore(store=store)

    live = {}  # pid -> ("assumed"|"committed", req)
    for kind, pid, size in ops:
        if kind == "check_assume" and pid not in live:
            req = AllocRequest(pod_name=f"q{pid}", namespace="fuzz",
                               request=Resource(size * 100.0,
                                                size * (10 << 30), 0.0))
            try:
                qs.check(req)
            except QuotaExceeded:
                continue
            qs.assume(req)
            live[pid] = ("assumed", req)
        elif kind == "commit" and live.get(pid, ("",))[0] == "assumed":
            qs.commit(live[pid][1])
            live[pid] = ("committed", live[pid][1])
        elif kind == "forget" and live.get(pid, ("",))[0] == "assumed":
            qs.forget(live[pid][1])
            del live[pid]
        elif kind == "release" and live.get(pid, ("",))[0] == "committed":
            qs.release(live[pid][1])
            del live[pid]

        u = qs.usage("fuzz")
        # the admitted set never exceeds the namespace total
        total = sum(r.request.tflops for _, r in live.values())
        assert total <= 1000.0 + 1e-9
        assert len(live) <= 6
        assert u.tflops <= 1000.0 + 1e-9

    for pid, (phase, req) in list(live.items()):
        qs.forget(req) if phase == "assumed" else qs.release(req)
    u = qs.usage("fuzz")
    assert u.tflops == 0 and u.vram == 0


@settings(max_examples=80, deadline=None)
@given(st.lists(
    st.tuples(st.sampled_from(["cluster", "node", "release"]),
              st.integers(min_value=0, max_value=30),
              st.integers(min_value=0, max_value=3)),
    min_size=1, max_size=120))
def test_port_allocator_never_double_assigns(ops):
    """Fuzzed assign/release: a port is never held by two pods at once
    and always comes from its documented range (reference
    portallocator.go:36-307 bitmaps)."""

    from tensor_fusion_amd.portallocator import PortAllocator, PortExhausted

    pa = PortAllocator()
    held = {}  # pod_key -> (kind, node, port)
    for kind, pid, nd in ops:
        key = f"ns/p{pid}"
        if kind == "release":
            pa.release(key)
            held.pop(key, None)
            continue
        if key in held:
            continue
        try:
            if kind == "cluster":
                port = pa.assign_cluster_port(key)
                assert 42000 <= port < 62000, port
            else:
                port = pa.assign_node_port(f"node-{nd}", key)
                assert 40000 <= port < 42000, port
        except PortExhausted:
            continue
        held[key] = (kind, nd, port)
        # no two pods hold the same port in the same scope
        seen = set()
        for k, (kk, n, p) in held.items():
            scope = ("cluster",) if kk == "cluster" else ("node", n)
            assert (scope, p) not in seen, (k, scope, p)
            seen.add((scope, p))
    for key in list(held):
        pa.release(key)
    assert pa.in_use() == 0


@settings(max_examples=200, deadline=None)
@given(st.lists(st.integers(min_value=0, max_value=7), min_size=1,
                max_size=8, unique=True),
       st.floats(min_value=0.1, max_value=100.0))
def test_cu_masks_cover_exactly_requested_cus(xcds, percent):
    """CU-mask composition invariants: XCD masks cover exactly 32 CUs per
    XCD with correct ids; percent masks round to the documented 1-CU
    granularity and never exceed 256."""

    from tensor_fusion_amd.allocator.partitioning import (cu_mask_for_percent,
                                                          cu_mask_for_xcds)

    mask = cu_mask_for_xcds(xcds)
    cus = set()
    for part in mask.split(","):
        lo, hi = (int(v) for v in part.split("-"))
        assert 0 <= lo <= hi < 256
        cus.update(range(lo, hi + 1))
    expect = set()
    for x in xcds:
        expect.update(range(32 * x, 32 * x + 32))
    assert cus == expect

    m2, count = cu_mask_for_percent(percent)
    assert 1 <= count <= 256
    assert abs(count - 256 * percent / 100.0) <= 1.0
    lo, hi = (int(v) for v in m2.split("-"))
    assert hi - lo + 1 == count


@settings(max_examples=60, deadline=None)
@given(st.lists(
    st.tuples(st.integers(min_value=0, max_value=19),   # pod id
              st.integers(min_value=1, max_value=7),    # vram tenths
              st.integers(min_value=0, max_value=2)),   # preferred node
    min_size=1, max_size=18))
def test_defrag_simulation_is_sound(placed):
    """Fuzzed fleets: when the defrag planner claims a candidate node can
    be drained, every evicted allocation must (a) actually live on a
    candidate and (b) be re-placed only on surviving nodes — and the
    plan must cover ALL allocations on the candidates (the reference's
    joint-placement simulation contract, gpupool_defrag.go:1095)."""

    from tensor_fusion_amd.api.store import Store
    from tensor_fusion_amd.controllers.defrag import DefragController

    a = GpuAllocator(store=None)
    for n in range(3):
        for i in range(4):
            g = GPU()
            g.meta.name = f"n{n}-g{i}"
            g.status.uuid = g.meta.name
            g.status.node = f"n{n}"
            g.status.capacity = Resource(CAP_T, CAP_V, 100.0)
            g.status.available = Resource(CAP_T, CAP_V, 100.0)
            a.upsert_gpu_for_testing(g)

    gpu_to_node = {g.meta.name: g.status.node for g in a.gpus()}
    for pid, vf, node_pref in placed:
        key = f"d/p{pid}"
        if any(k == key for k, _ in a.allocations_on(
                [f"n{n}" for n in range(3)])):
            continue
        req = AllocRequest(pod_name=f"p{pid}", namespace="d",
                           request=Resource(50.0, int(CAP_V * vf / 10), 5.0),
                           limit=Resource(CAP_T, CAP_V, 100.0))
        scores, _ = a.check_quota_and_filter(req)
        if not scores:
            continue
        node = f"n{node_pref}" if f"n{node_pref}" in scores \
            else max(scores, key=lambda n: scores[n].score)
        try:
            picked = a.pick_gpus(req, node)
            a.assume(req, picked)
            a.commit(req.pod_key)
            a.notify_bound(req.pod_key)
        except (AllocationError, KeyError):
            continue

    d = DefragController(store=Store(), allocator=a)
    for cand in (["n0"], ["n1"], ["n2"], ["n0", "n1"]):
        on_cand = {k for k, _ in a.allocations_on(cand)}
        plan = d.simulate(cand)
        if plan is None:
            continue
        assert set(plan.evict_pods) == on_cand
        for pod_key, gpus in plan.placements.items():
            assert pod_key in on_cand
            for g in gpus:
                assert gpu_to_node[g] not in cand, (pod_key, g)


@settings(max_examples=100, deadline=None)
@given(st.integers(min_value=2, max_value=6),
       st.lists(st.tuples(
           st.sampled_from(["arrive", "permit", "schedule", "reject"]),
           st.integers(min_value=0, max_value=7)),
           min_size=1, max_size=40))
def test_gang_all_or_nothing_invariants(min_members, ops):
    """Fuzzed gang lifecycle: Permit never releases a pod before quorum
    (waiting+scheduled ≥ min_members), and a rejection empties the
    waiting set atomically (reference gang/manager.go strict
    all-or-nothing :262/:1099)."""

    from tensor_fusion_amd.api.types import Pod
    from tensor_fusion_amd.gang.manager import GangManager

    gm = GangManager()
    pods = {}

    def mk(i):
        p = Pod()
        p.meta.name = f"g{i}"
        p.meta.namespace = "gang"
        p.meta.annotations[C.AnnoGangEnabled] = "true"
        p.meta.annotations[C.AnnoGangGroupKey] = "grp"
        p.meta.annotations[C.AnnoGangMinMembers] = str(min_members)
        return p

    for kind, i in ops:
        if kind == "arrive":
            pods[i] = mk(i)
            gm.register_pod(pods[i])
        elif kind == "permit" and i in pods:
            wait = gm.permit(pods[i])
            g = gm.groups.get("grp")
            if wait is None:
                # released immediately → quorum must actually be met
                assert g.quorum_now >= min_members, \
                    (g.quorum_now, min_members)
            else:
                assert wait > 0
        elif kind == "schedule" and i in pods:
            gm.mark_scheduled(pods[i])
        elif kind == "reject":
            gm.reject_group("grp")
            g = gm.groups.get("grp")
            if g is not None:
                assert not g.waiting  # atomic clear
                # a rejected group backs off: nothing admits until expiry
                assert gm.pre_enqueue(mk(99)) is not None


_ann_text = st.text(
    alphabet=st.characters(min_codepoint=32, max_codepoint=126), max_size=24)


@settings(max_examples=100, deadline=None)
@given(st.dictionaries(
    st.sampled_from([C.AnnoTflopsRequest, C.AnnoTflopsLimit,
                     C.AnnoVramRequest, C.AnnoVramLimit, C.AnnoGpuCount,
                     C.AnnoComputePercentRequest, C.AnnoQos,
                     C.AnnoIsolation, C.AnnoGangMinMembers,
                     C.AnnoGpuIndices, C.AnnoHostPort, C.AnnoGpuModel]),
    _ann_text, max_size=8),
    st.booleans())
def test_admission_boundary_never_500s_on_garbage(annotations, enabled):
    """Admission is the untrusted-input boundary: arbitrary annotation
    values (garbage numbers, empty strings, punctuation) must never
    crash the webhook endpoint — a malformed TF pod gets a clean
    allowed:False denial, a non-TF pod passes through untouched
    (reference pod_webhook.go Handle error responses)."""

    fastapi = pytest.importorskip("fastapi")
    from fastapi.testclient import TestClient

    from tensor_fusion_amd.api.store import Store
    from tensor_fusion_amd.server.webhook_server import create_webhook_app
    from tensor_fusion_amd.webhook import PodMutator

    labels = {C.LabelEnabled: "true"} if enabled else {}
    review = {"apiVersion": "admission.k8s.io/v1", "kind": "AdmissionReview",
              "request": {"uid": "u", "namespace": "default", "object": {
                  "metadata": {"name": "fz", "namespace": "default",
                               "labels": labels,
                               "annotations": dict(annotations)},
                  "spec": {"containers": [{"name": "main"}]}}}}
    client = TestClient(create_webhook_app(PodMutator(Store())))
    r = client.post("/mutate-v1-pod", json=review)
    assert r.status_code == 200  # never a 500
    resp = r.json()["response"]
    if not enabled:
        assert resp["allowed"] is True and not resp.get("patch")
    else:
        # either a clean denial (bad values) or a successful mutation
        if resp["allowed"]:
            assert resp.get("patch") or True
        else:
            assert "mutation failed" in resp["status"]["message"]


@settings(max_examples=150, deadline=None)
@given(st.floats(min_value=1.0, max_value=100.0),
       st.lists(st.floats(min_value=0.0, max_value=150.0),
                min_size=1, max_size=80),
       st.floats(min_value=0.05, max_value=2.0))
def test_erl_pid_rate_always_bounded_and_slew_limited(setpoint, utils, dt):
    """Fuzzed PID inputs (any utilization trace incl. >100% readings):
    the refill rate stays inside [min_rate, max_rate] and each step moves
    at most the configured slew — the stability contract the hypervisor
    loop depends on (reference quota_controller.go:321-376)."""

    from tensor_fusion_amd.hypervisor.erl import PidController

    c = PidController()
    p = c.p
    prev = c.state.rate
    for u in utils:
        rate = c.step(setpoint, u, dt)
        assert p.min_rate <= rate <= p.max_rate
        # per-step multiplicative slew clamp (unless pinned at a bound)
        if rate not in (p.min_rate, p.max_rate):
            assert rate <= prev * (1 + p.slew_up_percent / 100.0) + 1e-9
            assert rate >= prev * (1 - p.slew_down_percent / 100.0) - 1e-9
        prev = rate

    # convergence on an ideal linear plant: util = k * rate
    c2 = PidController()
    k = 0.02  # util% per token/s
    r = c2.state.rate
    for _ in range(400):
        r = c2.step(setpoint, k * r, 0.5)
    final_util = k * r
    if p.min_rate < r < p.max_rate:  # reachable setpoint
        assert abs(final_util - setpoint) <= \
            max(p.deadband_percent * 2, 0.12 * setpoint), \
            (setpoint, final_util)


@settings(max_examples=60, deadline=None)
@given(st.integers(min_value=2, max_value=8),
       st.integers(min_value=5, max_value=40))
def test_store_concurrent_patches_lose_no_updates(n_threads, n_incr):
    """The store's patch() retry loop under real thread contention:
    N threads × M increments on one object must all land (no lost
    updates through the RV-conflict path) and every event fires."""

    import threading

    from tensor_fusion_amd.api.store import Store
    from tensor_fusion_amd.api.types import GPUPool

    store = Store()
    p = GPUPool()
    p.meta.name = "ctr"
    store.create(p)
    events = []
    store.on_change("GPUPool", lambda e, o: events.append(e))

    def worker():
        for _ in range(n_incr):
            def _p(obj):
                obj.status.gpu_count += 1
            store.patch("GPUPool", "ctr", "", _p)

    threads = [threading.Thread(target=worker) for _ in range(n_threads)]
    for t in threads:
        t.start()
    for t in threads:
        t.join()
    got = store.get("GPUPool", "ctr")
    assert got.status.gpu_count == n_threads * n_incr
