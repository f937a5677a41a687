"""Wire-faithful in-process kube-apiserver for tests.

This build environment has no network, so BASELINE config 1's "kind
cluster" cannot be installed; this FastAPI app stands in for the
apiserver at the HTTP level the real one speaks: generic resource
storage with resourceVersions, label selectors, merge/json PATCH, the
status and binding subresources, chunked watch streams with replay and
410-too-old semantics, and mutating-admission dispatch (a registered
MutatingWebhookConfiguration with a `url` clientConfig is actually
invoked on pod CREATE and its JSONPatch applied — the reference's
webhook path, internal/webhook/v1/pod_webhook.go:84, runs over the real
AdmissionReview wire contract here).

K8sClient (client.py) does not special-case this server; everything it
does here works against a real apiserver.
"""
from __future__ import annotations

import base64
import copy
import json
import threading
import time
from collections import deque
from queue import Empty, Queue
from typing import Any, Dict, List, Optional, Tuple

from fastapi import FastAPI, Request, Response
from fastapi.responses import JSONResponse, StreamingResponse


def _now() -> str:
    return time.strftime("%Y-%m-%dT%H:%M:%SZ", time.gmtime())


def _match_selector(labels: Dict[str, str], selector: str) -> bool:
    if not selector:
        return True
    for term in selector.split(","):
        term = term.strip()
        if "!=" in term:
            k, v = term.split("!=", 1)
            if labels.get(k.strip()) == v.strip():
                return False
        elif "=" in term:
            k, v = term.split("=", 1)
            if labels.get(k.strip().rstrip("=")) != v.strip():
                return False
        elif term:
            if term not in labels:
                return False
    return True


def _json_patch(doc: Any, patch: List[dict]) -> Any:
    """Minimal RFC-6902 (add/replace/remove/test on object paths)."""

    doc = copy.deepcopy(doc)
    for op in patch:
        path = [p.replace("~1", "/").replace("~0", "~")
                for p in op["path"].split("/")[1:]]
        parent = doc
        for seg in path[:-1]:
            parent = parent[int(seg)] if isinstance(parent, list) else \
                parent.setdefault(seg, {})
        last = path[-1] if path else ""
        kind = op["op"]
        if kind in ("add", "replace"):
            if isinstance(parent, list):
                if last == "-":
                    parent.append(op["value"])
                elif kind == "add":
                    parent.insert(int(last), op["value"])
                else:
                    parent[int(last)] = op["value"]
            else:
                parent[last] = op["value"]
        elif kind == "remove":
            if isinstance(parent, list):
                parent.pop(int(last))
            else:
                parent.pop(last, None)
    return doc


def _merge_patch(doc: Any, patch: Any) -> Any:
    """RFC 7386 merge patch."""

    if not isinstance(patch, dict):
        return copy.deepcopy(patch)
    if not isinstance(doc, dict):
        doc = {}
    out = copy.deepcopy(doc)
    for k, v in patch.items():
        if v is None:
            out.pop(k, None)
        else:
            out[k] = _merge_patch(out.get(k), v)
    return out


class _ResourceTable:
    def __init__(self):
        self.objs: Dict[Tuple[str, str], dict] = {}  # (ns, name) -> obj
        self.watchers: List[Queue] = []
        self.history: deque = deque(maxlen=4096)  # (rv, type, obj)


class FakeApiServer:
    """The storage + semantics; `app` is the FastAPI wrapper."""

    def __init__(self):
        self._mu = threading.RLock()
        self._rv = 0
        self._tables: Dict[Tuple[str, str], _ResourceTable] = {}
        self._uid = 0
        self.app = self._build_app()

    # ------------------------------------------------------------ store

    def _table(self, group: str, resource: str) -> _ResourceTable:
        return self._tables.setdefault((group, resource), _ResourceTable())

    def _next_rv(self) -> str:
        self._rv += 1
        return str(self._rv)

    def _emit(self, tab: _ResourceTable, typ: str, obj: dict):
        rv = int(obj["metadata"]["resourceVersion"])
        tab.history.append((rv, typ, copy.deepcopy(obj)))
        for q in list(tab.watchers):
            q.put((typ, copy.deepcopy(obj)))

    def _store(self, group: str, resource: str, obj: dict,
               typ: str) -> dict:
        tab = self._table(group, resource)
        meta = obj.setdefault("metadata", {})
        key = (meta.get("namespace", ""), meta["name"])
        # deletion completes once a deleting object's finalizers empty
        # (real-apiserver garbage collection semantics)
        if typ == "MODIFIED" and meta.get("deletionTimestamp") \
                and not meta.get("finalizers"):
            tab.objs.pop(key, None)
            meta["resourceVersion"] = self._next_rv()
            self._emit(tab, "DELETED", obj)
            return copy.deepcopy(obj)
        meta["resourceVersion"] = self._next_rv()
        if typ == "ADDED":
            self._uid += 1
            meta.setdefault("uid", f"fake-uid-{self._uid}")
            meta.setdefault("creationTimestamp", _now())
        tab.objs[key] = copy.deepcopy(obj)
        self._emit(tab, typ, obj)
        return copy.deepcopy(obj)

    # -------------------------------------------------------- admission

    def _run_admission(self, obj: dict) -> dict:
        """Invoke registered mutating webhooks (url clientConfig) for pod
        creations, apply returned JSONPatches."""

        if obj.get("kind") != "Pod":
            return obj
        with self._mu:
            tab = self._tables.get(("admissionregistration.k8s.io",
                                    "mutatingwebhookconfigurations"))
            cfgs = [copy.deepcopy(o) for o in tab.objs.values()] if tab else []
        if not cfgs:
            return obj
        import requests
        for cfg in cfgs:
            for wh in cfg.get("webhooks", []):
                url = (wh.get("clientConfig") or {}).get("url")
                if not url:
                    continue
                sel = (wh.get("objectSelector") or {}).get("matchLabels") or {}
                labels = obj.get("metadata", {}).get("labels") or {}
                if any(labels.get(k) != v for k, v in sel.items()):
                    continue
                self._uid += 1
                review = {
                    "apiVersion": "admission.k8s.io/v1",
                    "kind": "AdmissionReview",
                    "request": {
                        "uid": f"adm-{self._uid}",
                        "kind": {"group": "", "version": "v1", "kind": "Pod"},
                        "namespace": obj["metadata"].get("namespace", ""),
                        "operation": "CREATE",
                        "object": obj,
                    },
                }
                try:
                    resp = requests.post(url, json=review, timeout=10,
                                         verify=False).json()
                except Exception:
                    continue  # failurePolicy: Ignore
                r = resp.get("response", {})
                if not r.get("allowed", True):
                    raise PermissionError(
                        r.get("status", {}).get("message", "denied"))
                if r.get("patch"):
                    patch = json.loads(base64.b64decode(r["patch"]))
                    obj = _json_patch(obj, patch)
        return obj

    # ------------------------------------------------------------- app

    def _build_app(self) -> FastAPI:
        app = FastAPI(title="fake-kube-apiserver")
        srv = self

        def parse(path: str) -> Optional[dict]:
            """Split an api path into group/resource/ns/name/sub."""

            parts = [p for p in path.split("/") if p]
            if not parts:
                return None
            if parts[0] == "api" and len(parts) >= 2 and parts[1] == "v1":
                group, rest = "", parts[2:]
            elif parts[0] == "apis" and len(parts) >= 3:
                group, rest = parts[1], parts[3:]
            else:
                return None
            ns = ""
            if rest and rest[0] == "namespaces" and len(rest) >= 3:
                ns, rest = rest[1], rest[2:]
            elif rest and rest[0] == "namespaces":
                rest = rest[0:1]  # list namespaces as a resource
            if not rest:
                return None
            resource = rest[0]
            name = rest[1] if len(rest) > 1 else ""
            sub = rest[2] if len(rest) > 2 else ""
            return {"group": group, "resource": resource, "ns": ns,
                    "name": name, "sub": sub}

        @app.get("/livez")
        def livez():
            return Response("ok")

        @app.api_route("/{full_path:path}",
                       methods=["GET", "POST", "PUT", "PATCH", "DELETE"])
        async def handle(full_path: str, request: Request):
            p = parse("/" + full_path)
            if p is None:
                return JSONResponse({"message": "not found"}, status_code=404)
            body = None
            if request.method in ("POST", "PUT", "PATCH"):
                body = json.loads(await request.body() or b"null")
            # dispatch on the threadpool: admission webhooks call back
            # into this server over HTTP, so the event loop must stay
            # free while a handler blocks on the webhook round trip
            from starlette.concurrency import run_in_threadpool
            return await run_in_threadpool(srv._dispatch, request, p, body)

        return app

    # --------------------------------------------------------- dispatch

    def _dispatch(self, request, p: dict, body):
        group, resource = p["group"], p["resource"]
        ns, name, sub = p["ns"], p["name"], p["sub"]
        method = request.method
        q = request.query_params
        if method == "POST" and not p["sub"] and isinstance(body, dict) \
                and body.get("kind") == "Pod":
            # admission OUTSIDE the store lock: the webhook may call back
            # into this apiserver (e.g. the mutator reads profiles)
            try:
                body = self._run_admission(body)
            except PermissionError as e:
                return JSONResponse({"message": str(e)}, status_code=400)
        with self._mu:
            tab = self._table(group, resource)
            if method == "GET" and not name:
                if q.get("watch") in ("true", "1"):
                    return self._watch_response(
                        tab, ns, q.get("labelSelector", ""),
                        q.get("resourceVersion", ""))
                items = [copy.deepcopy(o) for (ons, _), o in tab.objs.items()
                         if (not ns or ons == ns)
                         and _match_selector(
                             o.get("metadata", {}).get("labels") or {},
                             q.get("labelSelector", ""))]
                return JSONResponse({
                    "apiVersion": "v1", "kind": "List",
                    "metadata": {"resourceVersion": str(self._rv)},
                    "items": items})
            key = (ns, name)
            cur = tab.objs.get(key)
            if method == "GET":
                if cur is None:
                    return JSONResponse({"message": f"{resource} {name} not found"},
                                        status_code=404)
                return JSONResponse(copy.deepcopy(cur))
            if method == "POST" and sub == "binding":
                if cur is None:
                    return JSONResponse({"message": "pod not found"},
                                        status_code=404)
                node = (body.get("target") or {}).get("name", "")
                cur = copy.deepcopy(cur)
                cur.setdefault("spec", {})["nodeName"] = node
                self._store(group, resource, cur, "MODIFIED")
                return JSONResponse({"kind": "Status", "status": "Success"},
                                    status_code=201)
            if method == "POST":
                meta = body.setdefault("metadata", {})
                if not meta.get("name") and meta.get("generateName"):
                    self._uid += 1
                    meta["name"] = meta["generateName"] + f"{self._uid:05x}"
                if ns:
                    meta["namespace"] = ns
                k2 = (meta.get("namespace", ""), meta["name"])
                if k2 in tab.objs:
                    return JSONResponse({"message": "already exists"},
                                        status_code=409)
                out = self._store(group, resource, body, "ADDED")
                return JSONResponse(out, status_code=201)
            if method == "PUT":
                if cur is None:
                    return JSONResponse({"message": "not found"},
                                        status_code=404)
                rv = body.get("metadata", {}).get("resourceVersion")
                if rv and rv != cur["metadata"]["resourceVersion"]:
                    return JSONResponse({"message": "conflict"},
                                        status_code=409)
                if sub == "status":
                    nxt = copy.deepcopy(cur)
                    nxt["status"] = body.get("status", {})
                else:
                    nxt = body
                    if "status" in cur and "status" not in nxt:
                        nxt["status"] = cur["status"]
                nxt.setdefault("metadata", {})["namespace"] = ns
                nxt["metadata"]["name"] = name
                nxt["metadata"].setdefault("uid",
                                           cur["metadata"].get("uid", ""))
                nxt["metadata"].setdefault(
                    "creationTimestamp",
                    cur["metadata"].get("creationTimestamp", _now()))
                out = self._store(group, resource, nxt, "MODIFIED")
                return JSONResponse(out)
            if method == "PATCH":
                if cur is None:
                    return JSONResponse({"message": "not found"},
                                        status_code=404)
                ct = request.headers.get("content-type", "")
                if "json-patch" in ct:
                    nxt = _json_patch(cur, body)
                else:  # merge / strategic treated as merge
                    if sub == "status":
                        nxt = copy.deepcopy(cur)
                        nxt["status"] = _merge_patch(
                            cur.get("status"), body.get("status", body))
                    else:
                        nxt = _merge_patch(cur, body)
                nxt["metadata"]["name"] = name
                out = self._store(group, resource, nxt, "MODIFIED")
                return JSONResponse(out)
            if method == "DELETE":
                if cur is None:
                    return JSONResponse({"message": "not found"},
                                        status_code=404)
                if cur.get("metadata", {}).get("finalizers"):
                    # real-apiserver semantics: finalizers defer the
                    # delete — set deletionTimestamp, wait for a
                    # controller to strip them
                    nxt = copy.deepcopy(cur)
                    nxt["metadata"].setdefault(
                        "deletionTimestamp",
                        "1970-01-01T00:00:01Z")
                    out = self._store(group, resource, nxt, "MODIFIED")
                    return JSONResponse(out)
                del tab.objs[key]
                cur = copy.deepcopy(cur)
                cur["metadata"]["resourceVersion"] = self._next_rv()
                self._emit(tab, "DELETED", cur)
                return JSONResponse({"kind": "Status", "status": "Success"})
        return JSONResponse({"message": "bad request"}, status_code=400)

    # ------------------------------------------------------------ watch

    def _watch_response(self, tab: _ResourceTable, ns: str,
                        selector: str, rv: str):
        """Replay history after `rv`, then stream live events."""

        start_rv = int(rv) if rv else None
        replay: List[tuple] = []
        if start_rv is not None:
            oldest = tab.history[0][0] if tab.history else self._rv + 1
            if start_rv + 1 < oldest and start_rv < self._rv:
                # RV fell off the history window → 410 Gone (client relists)
                def gone():
                    yield json.dumps({
                        "type": "ERROR",
                        "object": {"kind": "Status", "code": 410,
                                   "reason": "Expired"}}) + "\n"
                return StreamingResponse(gone(),
                                         media_type="application/json")
            replay = [(t, o) for (erv, t, o) in tab.history
                      if erv > start_rv]
        q: Queue = Queue()
        tab.watchers.append(q)

        def match(o):
            m = o.get("metadata", {})
            if ns and m.get("namespace", "") != ns:
                return False
            return _match_selector(m.get("labels") or {}, selector)

        def gen():
            try:
                for typ, obj in replay:
                    if match(obj):
                        yield json.dumps({"type": typ, "object": obj}) + "\n"
                deadline = time.time() + 300
                while time.time() < deadline:
                    try:
                        typ, obj = q.get(timeout=1.0)
                    except Empty:
                        continue
                    if match(obj):
                        yield json.dumps({"type": typ, "object": obj}) + "\n"
            finally:
                try:
                    tab.watchers.remove(q)
                except ValueError:
                    pass

        return StreamingResponse(gen(), media_type="application/json")


def serve_in_thread(server: Optional[FakeApiServer] = None,
                    port: int = 0) -> Tuple["FakeApiServer", str, Any]:
    """Run the fake apiserver on 127.0.0.1:<port> in a daemon thread.
    Returns (server, base_url, uvicorn_server) — call .should_exit on the
    uvicorn server to stop."""

    import socket

    import uvicorn
    srv = server or FakeApiServer()
    if port == 0:
        with socket.socket() as sk:
            sk.bind(("127.0.0.1", 0))
            port = sk.getsockname()[1]
    async def _raise_threadpool(scope, receive, send):
        # every active watch stream occupies a threadpool slot (sync
        # generators run via iterate_in_threadpool); the anyio default
        # of 40 deadlocks multi-process informer tests
        import anyio.to_thread
        anyio.to_thread.current_default_thread_limiter().total_tokens = 256
        await srv.app(scope, receive, send)

    cfg = uvicorn.Config(_raise_threadpool, host="127.0.0.1", port=port,
                        log_level="error")
    us = uvicorn.Server(cfg)
    th = threading.Thread(target=us.run, daemon=True)
    th.start()
    base = f"http://127.0.0.1:{port}"
    for _ in range(200):
        import requests
        try:
            if requests.get(base + "/livez", timeout=1).ok:
                break
        except Exception:
            time.sleep(0.05)
    return srv, base, us
