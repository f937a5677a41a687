"""Operator HTTP API (:8080) — the client-facing lookup service.

Reference: internal/server/ + router/ (Gin on :8080, cmd/main.go:343-394):
`GET /connection` long-poll returns the TensorFusionConnection URL with
pod service-account JWT auth (router/connection.go:27-80), host-port and
index assignment (leader-proxied in the reference), and debug dumps of
allocator state (allocator_info.go) and node-scaler state.

Auth: bearer tokens are `<namespace>:<pod>:<signature>` where signature =
HMAC-SHA256(secret, ns:pod) — the embedded stand-in for validating pod SA
JWTs against the API server (TokenReview); namespace/pod are extracted
from the token exactly like the reference parses the JWT payload.
"""
from __future__ import annotations

import hashlib
import hmac
import time
from typing import Optional

from fastapi import FastAPI, Header, Query
from fastapi.responses import JSONResponse

from .. import constants as C
from ..api.store import Store


def make_token(secret: str, namespace: str, pod: str) -> str:
    sig = hmac.new(secret.encode(), f"{namespace}:{pod}".encode(),
                   hashlib.sha256).hexdigest()[:32]
    return f"{namespace}:{pod}:{sig}"


def parse_token(secret: str, token: str):
    try:
        namespace, pod, sig = token.split(":", 2)
    except ValueError:
        return None
    want = hmac.new(secret.encode(), f"{namespace}:{pod}".encode(),
                    hashlib.sha256).hexdigest()[:32]
    if not hmac.compare_digest(sig, want):
        return None
    return namespace, pod


def create_operator_app(store: Store, allocator=None, port_allocator=None,
                        index_allocator=None, secret: str = "tf-dev-secret",
                        long_poll_s: float = 10.0,
                        expander=None) -> FastAPI:
    app = FastAPI(title="tensor-fusion-operator")

    def _auth(authorization: Optional[str]):
        if not authorization or not authorization.startswith("Bearer "):
            return None
        return parse_token(secret, authorization[len("Bearer "):])

    @app.get("/connection")
    def connection(name: str = Query(...), namespace: str = Query("default"),
                   authorization: Optional[str] = Header(None)):
        ident = _auth(authorization)
        if ident is None:
            return JSONResponse({"error": "unauthorized"}, status_code=401)
        if ident[0] != namespace:
            return JSONResponse({"error": "namespace mismatch"},
                                status_code=403)
        deadline = time.time() + long_poll_s
        while True:
            conn = store.try_get("TensorFusionConnection", name, namespace)
            if conn is not None and conn.status.connection_url:
                return {"connectionURL": conn.status.connection_url,
                        "worker": conn.status.worker}
            if time.time() >= deadline:
                return JSONResponse({"error": "no connection yet"},
                                    status_code=404)
            time.sleep(0.1)

    @app.post("/assign-host-port")
    def assign_host_port(pod_name: str = Query(...),
                         namespace: str = Query("default"),
                         authorization: Optional[str] = Header(None)):
        if _auth(authorization) is None:
            return JSONResponse({"error": "unauthorized"}, status_code=401)
        if port_allocator is None:
            return JSONResponse({"error": "no port allocator"},
                                status_code=501)
        port = port_allocator.assign_cluster_port(f"{namespace}/{pod_name}")
        return {"hostPort": port}

    @app.post("/assign-index")
    def assign_index(pod_name: str = Query(...),
                     namespace: str = Query("default"),
                     authorization: Optional[str] = Header(None)):
        if _auth(authorization) is None:
            return JSONResponse({"error": "unauthorized"}, status_code=401)
        if index_allocator is None:
            return JSONResponse({"error": "no index allocator"},
                                status_code=501)
        idx = index_allocator.assign(f"{namespace}/{pod_name}")
        return {"index": idx}

    @app.get("/allocator-info")
    def allocator_info():
        if allocator is None:
            return {"gpus": []}
        out = []
        for g in allocator.gpus():
            s = g.status
            out.append({
                "name": g.meta.name, "uuid": s.uuid, "node": s.node,
                "pool": s.pool, "phase": s.phase,
                "capacity": {"tflops": s.capacity.tflops,
                             "vram": s.capacity.vram,
                             "computePercent": s.capacity.compute_percent},
                "available": {"tflops": s.available.tflops,
                              "vram": s.available.vram,
                              "computePercent": s.available.compute_percent},
                "runningApps": s.running_apps,
                "partitions": [p.partition_id
                               for p in s.allocated_partitions],
            })
        return {"gpus": out}

    @app.get("/node-scaler-info")
    def node_scaler_info():
        """Autoscaler-integration debug state (reference router
        node_scaler_info.go): in-flight provisioned node claims and
        pending unschedulable demand."""

        claims = []
        if expander is not None:
            for c in expander.in_flight_claims():
                claims.append({"name": c.meta.name,
                               "instanceType": c.instance_type,
                               "pool": c.pool,
                               "phase": c.status.phase})
        pending = []
        for pod in store.list("Pod"):
            if pod.meta.annotations.get(f"{C.Domain}/unschedulable"):
                pending.append(pod.meta.key)
        return {"inFlightClaims": claims, "pendingPods": pending}

    @app.get("/healthz")
    def healthz():
        return {"ok": True}

    return app
