"""Webhook HTTP micro-bench — QPS per code path through the real
AdmissionReview endpoint (reference: internal/webhook/v1/
pod_webhook_bench_test.go:92-251 benches HTTP QPS per code path)."""
import json
import time

import pytest

import tensor_fusion_amd.constants as C

fastapi = pytest.importorskip("fastapi")
from fastapi.testclient import TestClient  # noqa: E402

from tensor_fusion_amd.api.store import Store  # noqa: E402
from tensor_fusion_amd.server.webhook_server import create_webhook_app  # noqa: E402
from tensor_fusion_amd.webhook import PodMutator  # noqa: E402


def _review(pod: dict) -> dict:
    return {"apiVersion": "admission.k8s.io/v1", "kind": "AdmissionReview",
            "request": {"uid": "u-1", "namespace":
                        pod["metadata"].get("namespace", "default"),
                        "object": pod}}


def _tf_pod(i: int) -> dict:
    return {"metadata": {"name": f"app-{i}", "namespace": "default",
                         "labels": {C.LabelEnabled: "true"},
                         "annotations": {C.AnnoTflopsRequest: "100",
                                         C.AnnoVramRequest: str(8 << 30)}},
            "spec": {"containers": [{"name": "main", "image": "app:1",
                                     "env": [],
                                     "resources": {}}]}}


def _plain_pod(i: int) -> dict:
    return {"metadata": {"name": f"plain-{i}", "namespace": "default"},
            "spec": {"containers": [{"name": "main", "image": "app:1"}]}}


def _client() -> TestClient:
    app = create_webhook_app(PodMutator(Store()))
    return TestClient(app)


def _qps(client, make_pod, n=300) -> float:
    bodies = [json.dumps(_review(make_pod(i))) for i in range(n)]
    t0 = time.perf_counter()
    for b in bodies:
        r = client.post("/mutate-v1-pod", content=b,
                        headers={"content-type": "application/json"})
        assert r.status_code == 200
    return n / (time.perf_counter() - t0)


def test_webhook_qps_tf_pod_path():
    client = _client()
    # correctness of one response first
    r = client.post("/mutate-v1-pod", json=_review(_tf_pod(0)))
    resp = r.json()["response"]
    assert resp["allowed"] is True
    patch = json.loads(
        __import__("base64").b64decode(resp["patch"]).decode())
    assert any(p["path"] == "/metadata/annotations" for p in patch)
    qps = _qps(client, _tf_pod)
    print(f"\nwebhook tf-pod path: {qps:.0f} admissions/s")
    assert qps > 100, qps


def test_webhook_qps_passthrough_path():
    client = _client()
    r = client.post("/mutate-v1-pod", json=_review(_plain_pod(0)))
    resp = r.json()["response"]
    assert resp["allowed"] is True
    assert not resp.get("patch")  # non-TF pod: no mutation
    qps = _qps(client, _plain_pod)
    print(f"webhook passthrough path: {qps:.0f} admissions/s")
    assert qps > 150, qps


def test_multiprocess_webhook_serving():
    """serve_multiprocess: N forked acceptors on one shared TCP socket,
    app_factory called post-fork, Nagle disabled (a response must not
    take the 40 ms delayed-ACK stall)."""

    import json
    import os
    import signal
    import socket
    import time

    import requests

    from tensor_fusion_amd.api.store import Store
    from tensor_fusion_amd.server.webhook_server import (
        create_webhook_app, serve_multiprocess)
    from tensor_fusion_amd.webhook import PodMutator

    with socket.socket() as sk:
        sk.bind(("127.0.0.1", 0))
        port = sk.getsockname()[1]

    def factory():
        return create_webhook_app(PodMutator(Store()))

    lsock, pids = serve_multiprocess(None, port, workers=2,
                                     host="127.0.0.1",
                                     app_factory=factory)
    try:
        assert len(pids) == 2
        body = json.dumps({
            "apiVersion": "admission.k8s.io/v1", "kind": "AdmissionReview",
            "request": {"uid": "u1", "namespace": "default",
                        "object": {"metadata": {"name": "p",
                                                "namespace": "default"},
                                   "spec": {"containers": [
                                       {"name": "m", "image": "i"}]}}}})
        s = requests.Session()
        deadline = time.time() + 15
        r = None
        while time.time() < deadline:
            try:
                r = s.post(f"http://127.0.0.1:{port}/mutate-v1-pod",
                           data=body,
                           headers={"content-type": "application/json"},
                           timeout=2)
                break
            except requests.ConnectionError:
                time.sleep(0.2)
        assert r is not None and r.status_code == 200
        assert r.json()["response"]["allowed"] is True
        # Nagle check: a warm request must be far below the 40 ms
        # delayed-ACK floor
        lats = []
        for _ in range(10):
            t0 = time.perf_counter()
            s.post(f"http://127.0.0.1:{port}/mutate-v1-pod", data=body,
                   headers={"content-type": "application/json"},
                   timeout=2)
            lats.append(time.perf_counter() - t0)
        assert min(lats) < 0.035, f"nagle stall? {min(lats)*1e3:.1f}ms"
    finally:
        for pid in pids:
            os.kill(pid, signal.SIGTERM)
            os.waitpid(pid, 0)
        lsock.close()
