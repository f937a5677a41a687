"""Concurrency tests (reference: *_race_test.go — allocator two-phase
consistency under parallel scheduling, shm mutex, partition races)."""
import threading

import pytest

import tensor_fusion_amd.constants as C
from tensor_fusion_amd.allocator.gpuallocator import (AllocationError,
                                                      GpuAllocator)
from tensor_fusion_amd.api.store import Store
from tensor_fusion_amd.api.types import GPU, AllocRequest, Resource
from tensor_fusion_amd.quota.quota_store import QuotaExceeded


def fleet(n_nodes=4, per_node=8):
    a = GpuAllocator(store=None)
    for n in range(n_nodes):
        for i in range(per_node):
            g = GPU()
            g.meta.name = f"n{n}-g{i}"
            g.status.uuid = g.meta.name
            g.status.node = f"n{n}"
            g.status.capacity = Resource(2500.0, C.MI355X_VRAM_BYTES, 100.0)
            g.status.available = Resource(2500.0, C.MI355X_VRAM_BYTES, 100.0)
            a.upsert_gpu_for_testing(g)
    return a


def test_parallel_assume_commit_never_oversubscribes():
    """64 threads race to place 90%-VRAM pods on 32 GPUs: exactly 32 may
    win; availability never goes negative; released capacity returns."""

    a = fleet()
    won, lost = [], []
    barrier = threading.Barrier(64)

    def worker(i):
        req = AllocRequest(
            pod_name=f"p{i}", namespace="r",
            request=Resource(2000.0, int(0.9 * C.MI355X_VRAM_BYTES), 90.0),
            limit=Resource(2500.0, C.MI355X_VRAM_BYTES, 100.0))
        barrier.wait()
        try:
            scores, _ = a.check_quota_and_filter(req)
            for node in sorted(scores, key=lambda n: -scores[n].score):
                try:
                    picked = a.pick_gpus(req, node)
                    a.assume(req, picked)
                    a.commit(req.pod_key)
                    won.append(req.pod_key)
                    return
                except AllocationError:
                    continue
            lost.append(i)
        except (QuotaExceeded, AllocationError):
            lost.append(i)

    threads = [threading.Thread(target=worker, args=(i,)) for i in range(64)]
    for t in threads:
        t.start()
    for t in threads:
        t.join(timeout=60)
    assert len(won) == 32, f"{len(won)} placements on 32 GPUs"
    for g in a.gpus():
        assert not g.status.available.any_negative(), g.meta.name
    # full release restores capacity
    for key in won:
        a.dealloc(key)
    for g in a.gpus():
        assert g.status.available.vram == C.MI355X_VRAM_BYTES


def test_assume_rollback_storm_is_clean():
    a = fleet(n_nodes=1, per_node=2)

    def churn(i):
        for j in range(50):
            req = AllocRequest(
                pod_name=f"c{i}-{j}", namespace="r",
                request=Resource(100.0, 1 << 30, 5.0),
                limit=Resource(100.0, 1 << 30, 5.0))
            try:
                picked = a.pick_gpus(req, "n0")
                a.assume(req, picked)
                if j % 2:
                    a.commit(req.pod_key)
                    a.dealloc(req.pod_key)
                else:
                    a.rollback(req.pod_key)
            except AllocationError:
                continue

    threads = [threading.Thread(target=churn, args=(i,)) for i in range(8)]
    for t in threads:
        t.start()
    for t in threads:
        t.join(timeout=120)
    for g in a.gpus():
        av = g.status.available
        assert av.vram == C.MI355X_VRAM_BYTES, av
        assert abs(av.tflops - 2500.0) < 1e-6
        assert not g.status.running_apps


def test_concurrent_store_patch_and_informer_notify():
    """Store.patch under thread contention after the fast-copy rewrite:
    concurrent patches on the same object must all land (no lost
    updates), handlers observe monotonically increasing RVs, and the
    stored object never aliases a caller's return value."""

    import threading

    from tensor_fusion_amd.api.store import Store
    from tensor_fusion_amd.api.types import GPU

    store = Store()
    g = GPU()
    g.meta.name = "g0"
    store.create(g)

    rvs = []
    mu = threading.Lock()
    store.on_change("GPU", lambda e, o: (
        mu.__enter__(), rvs.append(o.meta.resource_version),
        mu.__exit__(None, None, None)))

    N_THREADS, N_EACH = 8, 50

    def worker(t):
        for i in range(N_EACH):
            def bump(obj, t=t, i=i):
                obj.meta.annotations[f"t{t}"] = str(i)
            out = store.patch("GPU", "g0", "", bump)
            # mutating the returned copy must not corrupt the store
            out.meta.annotations["rogue"] = "x"

    ts = [threading.Thread(target=worker, args=(t,))
          for t in range(N_THREADS)]
    for t in ts:
        t.start()
    for t in ts:
        t.join()

    final = store.get("GPU", "g0")
    assert "rogue" not in final.meta.annotations
    for t in range(N_THREADS):
        assert final.meta.annotations[f"t{t}"] == str(N_EACH - 1)
    assert len(rvs) == N_THREADS * N_EACH
    assert rvs == sorted(rvs), "handler RVs must be monotonic"


def test_percentile_recommender_under_concurrent_ingest():
    """TSDB ingest races the percentile recommender (reference
    percentile_recommender_race_test.go): recommendations stay within
    the observed value range and nothing throws."""

    import threading

    from tensor_fusion_amd.metrics.tsdb import TSDB

    db = TSDB(":memory:")
    stop = threading.Event()
    errs = []

    def ingest():
        i = 0
        while not stop.is_set():
            try:
                db.ingest_lines(
                    [f'worker_metrics,worker=w0 compute_percent='
                     f'{30 + (i % 40)} {1000000 + i}'])
            except Exception as e:  # pragma: no cover
                errs.append(e)
            i += 1

    reads = []

    def recommend():
        while not stop.is_set():
            try:
                rows = db.query("worker_metrics", "compute_percent")
                if rows:
                    vals = sorted(v for (_ts, v) in rows)
                    reads.append(vals[int(len(vals) * 0.9)])
            except Exception as e:  # pragma: no cover
                errs.append(e)

    ts = [threading.Thread(target=ingest) for _ in range(2)] + \
         [threading.Thread(target=recommend) for _ in range(2)]
    for t in ts:
        t.start()
    import time
    time.sleep(1.0)
    stop.set()
    for t in ts:
        t.join()
    assert not errs, errs[:3]
    assert reads and all(30 <= r <= 70 for r in reads)
