"""Metrics/TSDB/autoscaler/alert/operator-server/config tests."""
import os
import time

import pytest

import tensor_fusion_amd.constants as C
from tensor_fusion_amd.alert import AlertEvaluator, AlertRule
from tensor_fusion_amd.api.store import Store
from tensor_fusion_amd.api.types import (Resource, TensorFusionConnection,
                                         TensorFusionWorkload)
from tensor_fusion_amd.autoscaler import (Autoscaler, CronRecommender,
                                          DecayingHistogram,
                                          PercentileRecommender)
from tensor_fusion_amd.metrics import (MetricsRecorder, TSDB, WorkerMetrics,
                                       parse_influx_line)
from tensor_fusion_amd.server import create_operator_app, make_token


class TestMetrics:
    def test_influx_roundtrip(self):
        rec = MetricsRecorder()
        m = WorkerMetrics(workload="wl", worker="w0", namespace="d",
                          pool="p", qos=C.QosHigh, device_uuid="u0",
                          compute_percent=42.5, compute_tflops=600.0,
                          vram_bytes=8 << 30)
        rec.set_worker(m)
        lines = rec.encode_all(ts_ns=123)
        worker_lines = [l for l in lines if l.startswith("tf_worker_metrics")]
        assert len(worker_lines) == 1
        meas, tags, fields, ts = parse_influx_line(worker_lines[0])
        assert tags["workload"] == "wl" and tags["qos"] == "high"
        assert fields["compute_percent"] == 42.5
        assert ts == 123
        # billing: high-QoS pricing applied
        assert fields["raw_cost"] == pytest.approx(
            0.08 * 600 + 0.016 * 8, rel=1e-6)

    def test_rolling_file_and_tsdb_ingest(self, tmp_path):
        tsdb = TSDB()
        rec = MetricsRecorder(out_dir=str(tmp_path), tsdb=tsdb)
        rec.set_worker(WorkerMetrics(workload="wl", worker="w0",
                                     namespace="d", device_uuid="u0",
                                     compute_tflops=100.0))
        n = rec.flush()
        assert n >= 1
        assert os.path.exists(tmp_path / "metrics.log")
        pts = tsdb.query("tf_worker_metrics", "compute_tflops",
                         tags={"workload": "wl"})
        assert len(pts) == 1 and pts[0][1] == 100.0


class TestAutoscaler:
    def test_histogram_percentile(self):
        h = DecayingHistogram(half_life_s=1e9)  # no decay
        for v in [1.0] * 90 + [10.0] * 10:
            h.add(v, ts=h.ref_ts)
        assert h.percentile(0.5) == pytest.approx(1.0, rel=0.1)
        assert h.percentile(0.95) == pytest.approx(10.0, rel=0.15)

    def test_percentile_recommender_from_tsdb(self):
        tsdb = TSDB()
        rec = MetricsRecorder(tsdb=tsdb)
        for i in range(20):
            rec.set_worker(WorkerMetrics(
                workload="wl1", worker="w0", namespace="d", device_uuid="u0",
                compute_tflops=500.0, vram_bytes=40 << 30))
            rec.flush()
        store = Store()
        wl = TensorFusionWorkload()
        wl.meta.name = "wl1"
        wl.meta.namespace = "d"
        wl.profile.auto_scaling.enabled = True
        wl.profile.auto_scaling.recommender = "percentile"
        store.create(wl)
        a = Autoscaler(store, tsdb=tsdb, apply_in_place=False)
        assert a.tick() == 1
        got = store.get("TensorFusionWorkload", "wl1", "d")
        r = got.status.recommendation
        assert r is not None
        # p90 of constant 500 with 15% margin ∈ [500, 650]
        assert 450 <= r.resources.requests.tflops <= 700
        assert r.resources.requests.vram >= 35 << 30

    def test_cron_recommender_window(self):
        wl = TensorFusionWorkload()
        wl.profile.auto_scaling.cron_rules = [
            {"start": "00:00", "end": "23:59", "tflops": 777, "vram": 123}]
        r = CronRecommender().recommend(wl)
        assert r is not None and r.resources.requests.tflops == 777


class TestAlert:
    def test_rule_fires_and_resolves(self):
        tsdb = TSDB()
        rule = AlertRule(name="T", query="SELECT max(value) FROM points "
                         "WHERE field='x'", threshold=5.0, interval_s=0.0)
        ev = AlertEvaluator(tsdb, rules=[rule])
        tsdb.ingest_lines([f"m x=3.0 {time.time_ns()}"])
        assert ev.evaluate() == []  # below threshold, no transition
        tsdb.ingest_lines([f"m x=9.0 {time.time_ns()}"])
        changed = ev.evaluate()
        assert len(changed) == 1 and changed[0].firing
        assert ev.posted[-1]["status"] == "firing"


class TestOperatorServer:
    def test_connection_lookup_with_auth(self):
        from fastapi.testclient import TestClient
        store = Store()
        conn = TensorFusionConnection()
        conn.meta.name = "c1"
        conn.meta.namespace = "d"
        conn.status.connection_url = "native+10.0.0.9+8000+w-3"
        store.create(conn)
        app = create_operator_app(store, long_poll_s=0.2)
        c = TestClient(app)
        r = c.get("/connection", params={"name": "c1", "namespace": "d"})
        assert r.status_code == 401
        tok = make_token("tf-dev-secret", "d", "app-1")
        r = c.get("/connection", params={"name": "c1", "namespace": "d"},
                  headers={"Authorization": f"Bearer {tok}"})
        assert r.status_code == 200
        assert r.json()["connectionURL"] == "native+10.0.0.9+8000+w-3"
        # cross-namespace denied
        r = c.get("/connection", params={"name": "c1", "namespace": "other"},
                  headers={"Authorization": f"Bearer {tok}"})
        assert r.status_code == 403

    def test_bad_token_rejected(self):
        from fastapi.testclient import TestClient
        app = create_operator_app(Store(), long_poll_s=0.1)
        c = TestClient(app)
        r = c.get("/connection", params={"name": "x", "namespace": "d"},
                  headers={"Authorization": "Bearer d:app:deadbeef"})
        assert r.status_code == 401


class TestConfig:
    def test_hot_reload(self, tmp_path):
        from tensor_fusion_amd.config import ConfigWatcher
        p = tmp_path / "config.yaml"
        p.write_text("gpuFit:\n  vramWeight: 0.9\n")
        w = ConfigWatcher(str(p))
        assert w.config.gpu_fit.vram_weight == 0.9
        seen = []
        w.on_change(lambda cfg: seen.append(cfg.gpu_fit.vram_weight))
        time.sleep(0.01)
        os.utime(p, (time.time() + 5, time.time() + 5))
        p.write_text("gpuFit:\n  vramWeight: 0.4\n")
        os.utime(p, (time.time() + 10, time.time() + 10))
        assert w.reload() is True
        assert w.config.gpu_fit.vram_weight == 0.4
        assert seen == [0.4]

    def test_gpu_info_table(self, tmp_path):
        from tensor_fusion_amd.config import ConfigWatcher
        p = tmp_path / "config.yaml"
        p.write_text(
            "gpuInfo:\n- model: MI300X\n  fp16TFlops: 1300\n"
            "  vramBytes: 206158430208\n")
        w = ConfigWatcher(str(p))
        assert "MI300X" in w.config.gpu_info
        assert w.config.gpu_info["MI300X"].fp16_tflops == 1300
        assert "MI355X" in w.config.gpu_info  # default retained


class TestWebhookServer:
    def test_admission_review_roundtrip(self):
        import base64
        import json as _json

        from fastapi.testclient import TestClient

        from tensor_fusion_amd.server.webhook_server import create_webhook_app
        from tensor_fusion_amd.webhook import PodMutator

        store = Store()
        app = create_webhook_app(PodMutator(store))
        c = TestClient(app)
        review = {
            "apiVersion": "admission.k8s.io/v1",
            "kind": "AdmissionReview",
            "request": {
                "uid": "u-1",
                "object": {
                    "metadata": {"name": "app-1", "namespace": "default",
                                 "labels": {C.LabelEnabled: "true"},
                                 "annotations": {
                                     C.AnnoTflopsRequest: "500",
                                     C.AnnoVramRequest: "16Gi"}},
                    "spec": {"containers": [{"name": "main",
                                             "image": "app:1"}]},
                },
            },
        }
        r = c.post("/mutate-v1-pod", json=review)
        assert r.status_code == 200
        resp = r.json()["response"]
        assert resp["allowed"] is True
        patch = _json.loads(base64.b64decode(resp["patch"]))
        paths = {p["path"] for p in patch}
        assert "/spec/containers/0/env" in paths
        # workload CR was created by the same mutator core
        assert store.get("TensorFusionWorkload", "app-1-wl",
                         "default") is not None

    def test_non_tf_pod_untouched(self):
        from fastapi.testclient import TestClient

        from tensor_fusion_amd.server.webhook_server import create_webhook_app
        from tensor_fusion_amd.webhook import PodMutator

        app = create_webhook_app(PodMutator(Store()))
        c = TestClient(app)
        review = {"request": {"uid": "u-2", "object": {
            "metadata": {"name": "plain", "namespace": "d"},
            "spec": {"containers": [{"name": "c"}]}}}}
        r = c.post("/mutate-v1-pod", json=review)
        resp = r.json()["response"]
        assert resp["allowed"] is True
        assert "patch" not in resp


class TestConfigAlertRules:
    def test_rules_hot_reload_into_evaluator(self, tmp_path):
        from tensor_fusion_amd.operator import build_operator
        p = tmp_path / "config.yaml"
        p.write_text(
            "alertRules:\n"
            "- name: CustomHighUtil\n"
            "  query: \"SELECT max(value) FROM points WHERE field='u'\"\n"
            "  threshold: 90\n"
            "  severity: critical\n")
        op = build_operator(config_path=str(p))
        names = {r.name for r in op.alerts.rules}
        assert "CustomHighUtil" in names
        assert "PoolVramSaturation" in names  # defaults retained


def test_tui_renders_device_and_worker_frame():
    """TUI frame rendering from hypervisor API payloads (reference
    pkg/hypervisor/tui device/worker views) — no HTTP, pure render."""

    from rich.console import Console

    from tensor_fusion_amd.tui.app import UiState, build_frame

    devices = [{"index": 0, "uuid": "GPU-abc123", "vram_used": 24 << 30,
                "vram_total": 288 << 30, "busy_percent": 42.0,
                "compute_units": 256, "worker_count": 2}]
    workers = [{"namespace": "default", "pod": "w-0", "qos": "medium",
                "isolation": "soft",
                "limits": {"vram": 8 << 30, "compute_percent": 25},
                "usage": {"vram": 2 << 30, "erl_rate": 120.5,
                          "block_ns": 3_000_000},
                "heartbeat_ts": 0}]
    console = Console(record=True, width=120)
    console.print(build_frame(UiState(view="devices", devices=devices,
                                      workers=workers)))
    console.print(build_frame(UiState(view="workers", devices=devices,
                                      workers=workers)))
    text = console.export_text()
    assert "GPU-abc123"[:10] in text
    assert "288" in text and "24.0" in text
    assert "default/w-0" in text
    assert "medium" in text


def test_node_scaler_info_route():
    """GET /node-scaler-info exposes in-flight provisioning claims
    (reference router/node_scaler_info.go)."""

    from tensor_fusion_amd.api.types import GPUNodeClaim
    from tensor_fusion_amd.scheduler.expander import NodeExpander

    store = Store()
    c = GPUNodeClaim()
    c.meta.name = "claim-1"
    c.pool = "pool-a"
    c.instance_type = "mi355x.8x"
    store.create(c)
    app = create_operator_app(store, expander=NodeExpander(store))
    from fastapi.testclient import TestClient
    r = TestClient(app).get("/node-scaler-info")
    assert r.status_code == 200
    data = r.json()
    assert data["inFlightClaims"][0]["name"] == "claim-1"
    assert data["inFlightClaims"][0]["instanceType"] == "mi355x.8x"


def test_metrics_remote_sink_ships_and_backlogs():
    """External pipeline (reference: rolling file → Vector → GreptimeDB
    influx write endpoint): the recorder POSTs the same line protocol
    directly, with a bounded backlog across sink outages."""

    import threading

    received = []
    fail = {"on": True}
    from fastapi import FastAPI, Request, Response
    app = FastAPI()

    @app.post("/v1/influxdb/write")
    async def write(req: Request):
        if fail["on"]:
            return Response(status_code=503)
        received.append((await req.body()).decode())
        return Response(status_code=204)

    import socket
    import time

    import requests
    import uvicorn
    with socket.socket() as sk:
        sk.bind(("127.0.0.1", 0))
        port = sk.getsockname()[1]
    srv = uvicorn.Server(uvicorn.Config(app, host="127.0.0.1", port=port,
                                        log_level="error"))
    threading.Thread(target=srv.run, daemon=True).start()
    for _ in range(100):
        try:
            requests.get(f"http://127.0.0.1:{port}/x", timeout=1)
            break
        except Exception:
            time.sleep(0.02)
    try:
        from tensor_fusion_amd.metrics.recorder import (MetricsRecorder,
                                                        WorkerMetrics)
        rec = MetricsRecorder(
            remote_url=f"http://127.0.0.1:{port}/v1/influxdb/write",
            remote_auth="token abc")
        rec.set_worker(WorkerMetrics(namespace="ns", worker="w1",
                                     workload="wl", qos="medium",
                                     compute_tflops=10.0,
                                     vram_bytes=1 << 30))
        # sink down: flush backlogs, doesn't raise
        rec.flush()
        assert rec.remote_errors == 1 and not received
        # sink recovers: backlog + new lines ship together
        fail["on"] = False
        rec.flush()
        assert rec.remote_posts == 1
        assert received and "tf_worker_metrics" in received[0]
        assert received[0].count("tf_worker_metrics") >= 2  # backlog shipped
    finally:
        srv.should_exit = True


def test_recommendation_applied_at_admission():
    """Autoscaler feedback loop closes through the webhook (reference
    pod_webhook :349): a workload's status.recommendation resizes NEW
    replicas admitted without explicit resource annotations; explicit
    annotations always win."""

    import tensor_fusion_amd.constants as C
    from tensor_fusion_amd.api.store import Store
    from tensor_fusion_amd.api.types import (Pod, Recommendation,
                                             Requirements, Resource,
                                             TensorFusionWorkload)
    from tensor_fusion_amd.webhook import PodMutator

    store = Store()
    wl = TensorFusionWorkload()
    wl.meta.name = "app-wl"
    wl.meta.namespace = "default"
    wl.status.recommendation = Recommendation(
        resources=Requirements(requests=Resource(400.0, 32 << 30, 0),
                               limits=Resource(400.0, 32 << 30, 0)),
        reason="percentile")
    store.create(wl)
    mut = PodMutator(store)

    pod = Pod()
    pod.meta.name = "app-1"
    pod.meta.namespace = "default"
    pod.meta.labels = {C.LabelEnabled: "true", C.LabelWorkload: "app-wl"}
    prof = mut.parse(pod)
    assert prof.resources.requests.tflops == 400.0
    assert prof.resources.requests.vram == 32 << 30

    # explicit annotations pin the size — recommendation must not win
    pod2 = Pod()
    pod2.meta.name = "app-2"
    pod2.meta.namespace = "default"
    pod2.meta.labels = {C.LabelEnabled: "true", C.LabelWorkload: "app-wl"}
    pod2.meta.annotations = {C.AnnoTflopsRequest: "100",
                             C.AnnoVramRequest: str(8 << 30)}
    prof2 = mut.parse(pod2)
    assert prof2.resources.requests.tflops == 100.0
    assert prof2.resources.requests.vram == 8 << 30


def test_assign_port_and_index_routes():
    """Operator HTTP routes /assign-host-port and /assign-index
    (reference router assign_host_port.go / assign_index.go):
    authorized callers get monotonically distinct assignments; missing
    auth is 401."""

    from fastapi.testclient import TestClient

    from tensor_fusion_amd.operator import build_operator
    from tensor_fusion_amd.server import create_operator_app
    from tensor_fusion_amd.server.operator_server import make_token

    op = build_operator()
    app = create_operator_app(op.store, allocator=op.allocator,
                              port_allocator=op.port_allocator,
                              index_allocator=op.index_allocator)
    c = TestClient(app)
    assert c.post("/assign-host-port?pod_name=p1").status_code == 401

    hdr = {"Authorization": "Bearer " +
           make_token("tf-dev-secret", "default", "p1")}
    p1 = c.post("/assign-host-port?pod_name=p1", headers=hdr).json()
    p2 = c.post("/assign-host-port?pod_name=p2", headers=hdr).json()
    assert p1["hostPort"] != p2["hostPort"]
    assert 42000 <= p1["hostPort"] < 62000

    i1 = c.post("/assign-index?pod_name=p1", headers=hdr).json()
    i2 = c.post("/assign-index?pod_name=p2", headers=hdr).json()
    assert i1["index"] != i2["index"]
    assert 1 <= i1["index"] <= 32
    # idempotent per pod (re-admission must not leak a second slot)
    assert c.post("/assign-index?pod_name=p1",
                  headers=hdr).json()["index"] == i1["index"]
