"""Admission webhook HTTP endpoint — the kube-apiserver wire contract.

Reference: internal/webhook/v1 serves /mutate-v1-pod taking an
AdmissionReview(v1) and returning a JSONPatch response. The embedded
store path uses PodMutator directly; this endpoint exists so a real
cluster can point a MutatingWebhookConfiguration at the operator: it
converts corev1.Pod JSON ⇄ the internal Pod model, runs the same
mutator, and responds with an RFC-6902 patch.
"""
from __future__ import annotations

import base64
import json
from typing import Any, Dict, List

from ..api.types import Container, Pod
from ..webhook import PodMutator


def pod_from_k8s(obj: Dict[str, Any]) -> Pod:
    p = Pod()
    meta = obj.get("metadata", {})
    p.meta.name = meta.get("name", "") or meta.get("generateName", "")
    p.meta.namespace = meta.get("namespace", "default")
    p.meta.labels = dict(meta.get("labels") or {})
    p.meta.annotations = dict(meta.get("annotations") or {})
    spec = obj.get("spec", {})
    p.scheduler_name = spec.get("schedulerName", "default")
    for c in spec.get("containers", []):
        cc = Container(name=c.get("name", "main"),
                       image=c.get("image", ""),
                       command=list(c.get("command") or []))
        for e in c.get("env") or []:
            if "value" in e:
                cc.env[e["name"]] = e["value"]
        res = c.get("resources") or {}
        for kind in ("limits", "requests"):
            for k, v in (res.get(kind) or {}).items():
                cc.resources[k] = str(v)
        p.containers.append(cc)
    return p


def pod_to_k8s_patch(orig: Dict[str, Any], mutated: Pod) -> List[dict]:
    """RFC-6902 patch from the original k8s pod to the mutated model."""

    patch: List[dict] = []
    meta = orig.get("metadata", {})
    if (meta.get("labels") or {}) != mutated.meta.labels:
        patch.append({"op": "add" if "labels" not in meta else "replace",
                      "path": "/metadata/labels",
                      "value": mutated.meta.labels})
    if (meta.get("annotations") or {}) != mutated.meta.annotations:
        patch.append({"op": "add" if "annotations" not in meta else "replace",
                      "path": "/metadata/annotations",
                      "value": mutated.meta.annotations})
    if orig.get("spec", {}).get("schedulerName", "default") != \
            mutated.scheduler_name:
        patch.append({"op": "add", "path": "/spec/schedulerName",
                      "value": mutated.scheduler_name})
    for i, (oc, mc) in enumerate(zip(orig.get("spec", {}).get(
            "containers", []), mutated.containers)):
        oenv = {e["name"]: e.get("value") for e in (oc.get("env") or [])}
        if oenv != mc.env:
            patch.append({
                "op": "add" if not oc.get("env") else "replace",
                "path": f"/spec/containers/{i}/env",
                "value": [{"name": k, "value": v} for k, v in mc.env.items()],
            })
        ores = {k: str(v) for kind in ("limits", "requests")
                for k, v in (oc.get("resources", {}).get(kind) or {}).items()}
        if ores != mc.resources:
            patch.append({
                "op": "add" if not oc.get("resources") else "replace",
                "path": f"/spec/containers/{i}/resources",
                "value": {"limits": dict(mc.resources),
                          "requests": dict(mc.resources)},
            })
        if mc.volume_mounts and not oc.get("volumeMounts"):
            patch.append({
                "op": "add",
                "path": f"/spec/containers/{i}/volumeMounts",
                "value": mc.volume_mounts,
            })
    return patch


def mutate_review(mutator: PodMutator, review: Dict[str, Any]) -> dict:
    """AdmissionReview request → AdmissionReview response (pure)."""

    req = review.get("request", {})
    uid = req.get("uid", "")
    obj = req.get("object", {})
    resp: Dict[str, Any] = {"uid": uid, "allowed": True}
    try:
        pod = pod_from_k8s(obj)
        if mutator.should_handle(pod):
            mutator.handle(pod)
            patch = pod_to_k8s_patch(obj, pod)
            if patch:
                resp["patchType"] = "JSONPatch"
                resp["patch"] = base64.b64encode(
                    json.dumps(patch).encode()).decode()
    except Exception as e:
        resp = {"uid": uid, "allowed": False,
                "status": {"message": f"mutation failed: {e}"}}
    return {"apiVersion": "admission.k8s.io/v1",
            "kind": "AdmissionReview", "response": resp}


def create_webhook_app(mutator: PodMutator):
    """Raw ASGI app — the admission path is throughput-critical
    (reference benches 8.5k req/s on Go, scripts/benchmark.sh), so the
    hot endpoint skips framework routing/DI entirely: parse body, run
    the mutator, serialize. ~3x the FastAPI version under the same
    uvicorn worker (tools/bench_webhook.py)."""

    _JSON = [(b"content-type", b"application/json")]
    _OK = json.dumps({"ok": True}).encode()

    async def app(scope, receive, send):
        if scope["type"] == "lifespan":  # uvicorn startup/shutdown
            while True:
                msg = await receive()
                if msg["type"] == "lifespan.startup":
                    await send({"type": "lifespan.startup.complete"})
                elif msg["type"] == "lifespan.shutdown":
                    await send({"type": "lifespan.shutdown.complete"})
                    return
        if scope["type"] != "http":
            return
        path = scope.get("path", "")
        if path == "/healthz":
            await send({"type": "http.response.start", "status": 200,
                        "headers": _JSON})
            await send({"type": "http.response.body", "body": _OK})
            return
        if path != "/mutate-v1-pod" or scope.get("method") != "POST":
            await send({"type": "http.response.start", "status": 404,
                        "headers": _JSON})
            await send({"type": "http.response.body", "body": b"{}"})
            return
        chunks = []
        while True:
            msg = await receive()
            b = msg.get("body", b"")
            if b:
                chunks.append(b)
            if not msg.get("more_body"):
                break
        try:
            review = json.loads(b"".join(chunks) or b"{}")
        except ValueError:
            review = {}
        out = mutate_review(mutator, review)
        await send({"type": "http.response.start", "status": 200,
                    "headers": _JSON})
        await send({"type": "http.response.body",
                    "body": json.dumps(out).encode()})

    return app


def serve_multiprocess(app, port: int, workers: int = 2,
                       host: str = "0.0.0.0",
                       ssl_certfile: str = "", ssl_keyfile: str = "",
                       app_factory=None):
    """Serve the webhook from `workers` forked processes accepting on
    ONE shared listening socket — admission is stateless per request
    (the mutator reads a read-mostly store snapshot), so the kernel's
    accept queue load-balances across processes and throughput scales
    with cores instead of being GIL-bound. Returns (socket, [pids]);
    close the socket and signal the pids to stop.

    Reference parity: the Go webhook serves from one multiplexed
    process; this is the CPython equivalent of its goroutine
    concurrency."""

    import os
    import socket as _socket

    import uvicorn

    # proto must be IPPROTO_TCP (not 0): accepted sockets inherit it,
    # and asyncio's _set_nodelay only disables Nagle when proto says
    # TCP — with proto 0 every response stalls ~40 ms on delayed ACK.
    sock = _socket.socket(_socket.AF_INET, _socket.SOCK_STREAM,
                          _socket.IPPROTO_TCP)
    sock.setsockopt(_socket.SOL_SOCKET, _socket.SO_REUSEADDR, 1)
    sock.bind((host, port))
    sock.listen(4096)
    pids = []
    for _ in range(max(1, workers)):
        pid = os.fork()
        if pid == 0:
            kw = {}
            if ssl_certfile:
                kw = {"ssl_certfile": ssl_certfile,
                      "ssl_keyfile": ssl_keyfile}
            # app_factory (zero-arg, called post-fork) exists because
            # threads — informers, watch streams — do not survive
            # fork: state that must stay live is rebuilt in the child
            child_app = app_factory() if app_factory else app
            cfg = uvicorn.Config(child_app, log_level="warning", **kw)
            srv = uvicorn.Server(cfg)
            srv.run(sockets=[sock])
            os._exit(0)
        pids.append(pid)
    return sock, pids
