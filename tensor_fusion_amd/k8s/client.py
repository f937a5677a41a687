"""Minimal Kubernetes REST client.

No client-go port and no third-party SDK: the operator talks to the
apiserver over plain REST (requests.Session), which is all the reference's
controller-runtime manager ultimately does. Supports kubeconfig and
in-cluster auth, CRUD + PATCH (merge / json-patch), the status and binding
subresources, label selectors, and chunked watch streams.

Reference anchor: cmd/main.go:131-297 builds a manager against the
cluster; internal/gpuallocator syncs GPU CR status back with patches
(gpuallocator.go:2157-2621).
"""
from __future__ import annotations

import json
import os
import threading
from typing import Any, Callable, Dict, Iterator, List, Optional, Tuple

import requests

import yaml

from .serde import CLUSTER_SCOPED, GROUP, PLURALS, VERSION

CORE_KINDS = {"Pod": "pods", "Node": "nodes", "Namespace": "namespaces",
              "Event": "events", "ConfigMap": "configmaps",
              "Secret": "secrets", "Service": "services"}


class ApiError(Exception):
    def __init__(self, status: int, reason: str, body: str = ""):
        super().__init__(f"{status} {reason}: {body[:300]}")
        self.status = status
        self.reason = reason
        self.body = body

    @property
    def not_found(self) -> bool:
        return self.status == 404

    @property
    def conflict(self) -> bool:
        return self.status == 409

    @property
    def gone(self) -> bool:
        return self.status == 410


class K8sClient:
    def __init__(self, base_url: str, token: str = "",
                 ca_file: Optional[str] = None, verify: bool = True,
                 timeout: float = 30.0):
        self.base_url = base_url.rstrip("/")
        self.timeout = timeout
        self._s = requests.Session()
        if token:
            self._s.headers["Authorization"] = f"Bearer {token}"
        self._s.verify = ca_file if (ca_file and verify) else verify
        self._local = threading.local()

    # ------------------------------------------------------ construction

    @classmethod
    def from_kubeconfig(cls, path: Optional[str] = None,
                        context: Optional[str] = None) -> "K8sClient":
        path = path or os.environ.get(
            "KUBECONFIG", os.path.expanduser("~/.kube/config"))
        with open(path) as f:
            cfg = yaml.safe_load(f)
        ctx_name = context or cfg.get("current-context")
        ctx = next(c["context"] for c in cfg["contexts"]
                   if c["name"] == ctx_name)
        cluster = next(c["cluster"] for c in cfg["clusters"]
                       if c["name"] == ctx["cluster"])
        user = next(u["user"] for u in cfg["users"]
                    if u["name"] == ctx["user"])
        token = user.get("token", "")
        verify: Any = not cluster.get("insecure-skip-tls-verify", False)
        ca = cluster.get("certificate-authority")
        return cls(cluster["server"], token=token, ca_file=ca, verify=verify)

    @classmethod
    def in_cluster(cls) -> "K8sClient":
        sa = "/var/run/secrets/kubernetes.io/serviceaccount"
        host = os.environ["KUBERNETES_SERVICE_HOST"]
        port = os.environ.get("KUBERNETES_SERVICE_PORT", "443")
        with open(f"{sa}/token") as f:
            token = f.read().strip()
        return cls(f"https://{host}:{port}", token=token,
                   ca_file=f"{sa}/ca.crt")

    @classmethod
    def auto(cls) -> "K8sClient":
        """in-cluster if the SA mount exists, else kubeconfig, else the
        TF_K8S_URL env (plain http, e.g. the fake apiserver)."""

        if os.path.exists(
                "/var/run/secrets/kubernetes.io/serviceaccount/token"):
            return cls.in_cluster()
        url = os.environ.get("TF_K8S_URL")
        if url:
            return cls(url)
        return cls.from_kubeconfig()

    # ------------------------------------------------------------ paths

    def _path(self, kind: str, namespace: str = "", name: str = "",
              subresource: str = "") -> str:
        if kind in CORE_KINDS:
            base = "/api/v1"
            plural = CORE_KINDS[kind]
            namespaced = kind not in ("Node", "Namespace")
        elif kind == "Lease":
            base = "/apis/coordination.k8s.io/v1"
            plural, namespaced = "leases", True
        elif kind == "CustomResourceDefinition":
            base = "/apis/apiextensions.k8s.io/v1"
            plural, namespaced = "customresourcedefinitions", False
        elif kind == "MutatingWebhookConfiguration":
            base = "/apis/admissionregistration.k8s.io/v1"
            plural, namespaced = "mutatingwebhookconfigurations", False
        else:
            base = f"/apis/{GROUP}/{VERSION}"
            plural = PLURALS.get(kind, kind.lower() + "s")
            namespaced = kind not in CLUSTER_SCOPED
        p = base
        if namespaced and namespace:
            p += f"/namespaces/{namespace}"
        p += f"/{plural}"
        if name:
            p += f"/{name}"
        if subresource:
            p += f"/{subresource}"
        return p

    # ------------------------------------------------------------- HTTP

    def _req(self, method: str, path: str, params: Optional[dict] = None,
             body: Any = None, content_type: str = "application/json",
             stream: bool = False, timeout: Optional[float] = None
             ) -> requests.Response:
        url = self.base_url + path
        headers = {"Content-Type": content_type} if body is not None else {}
        r = self._s.request(method, url, params=params,
                            data=json.dumps(body) if body is not None else None,
                            headers=headers, stream=stream,
                            timeout=timeout or self.timeout)
        if r.status_code >= 400:
            raise ApiError(r.status_code, r.reason, r.text)
        return r

    # -------------------------------------------------------------- API

    def get(self, kind: str, name: str, namespace: str = "") -> dict:
        return self._req("GET", self._path(kind, namespace, name)).json()

    def try_get(self, kind: str, name: str,
                namespace: str = "") -> Optional[dict]:
        try:
            return self.get(kind, name, namespace)
        except ApiError as e:
            if e.not_found:
                return None
            raise

    def list(self, kind: str, namespace: str = "",
             label_selector: str = "",
             resource_version: str = "") -> dict:
        params = {}
        if label_selector:
            params["labelSelector"] = label_selector
        if resource_version:
            params["resourceVersion"] = resource_version
        return self._req("GET", self._path(kind, namespace),
                         params=params).json()

    def list_items(self, kind: str, namespace: str = "",
                   label_selector: str = "") -> List[dict]:
        return self.list(kind, namespace, label_selector).get("items", [])

    def create(self, obj: dict, namespace: str = "") -> dict:
        kind = obj["kind"]
        ns = namespace or obj.get("metadata", {}).get("namespace", "")
        return self._req("POST", self._path(kind, ns), body=obj).json()

    def update(self, obj: dict) -> dict:
        kind = obj["kind"]
        meta = obj.get("metadata", {})
        return self._req(
            "PUT", self._path(kind, meta.get("namespace", ""), meta["name"]),
            body=obj).json()

    def update_status(self, obj: dict) -> dict:
        kind = obj["kind"]
        meta = obj.get("metadata", {})
        return self._req(
            "PUT", self._path(kind, meta.get("namespace", ""), meta["name"],
                              "status"),
            body=obj).json()

    def patch(self, kind: str, name: str, patch: Any, namespace: str = "",
              patch_type: str = "merge",
              subresource: str = "") -> dict:
        ct = {"merge": "application/merge-patch+json",
              "json": "application/json-patch+json",
              "strategic": "application/strategic-merge-patch+json"}[patch_type]
        return self._req("PATCH",
                         self._path(kind, namespace, name, subresource),
                         body=patch, content_type=ct).json()

    def delete(self, kind: str, name: str, namespace: str = "") -> None:
        self._req("DELETE", self._path(kind, namespace, name))

    def bind_pod(self, name: str, namespace: str, node: str) -> None:
        """POST pods/{name}/binding — how a scheduler binds (kube API)."""

        body = {"apiVersion": "v1", "kind": "Binding",
                "metadata": {"name": name, "namespace": namespace},
                "target": {"apiVersion": "v1", "kind": "Node",
                           "name": node}}
        self._req("POST", self._path("Pod", namespace, name, "binding"),
                  body=body)

    # ------------------------------------------------------------ watch

    def watch(self, kind: str, namespace: str = "",
              resource_version: str = "", label_selector: str = "",
              timeout_s: float = 300.0) -> Iterator[Tuple[str, dict]]:
        """Yield (event_type, object) from a chunked watch stream until the
        server closes it. Raises ApiError(410) when the RV is too old —
        callers relist (informer.py does)."""

        params = {"watch": "true"}
        if resource_version:
            params["resourceVersion"] = resource_version
        if label_selector:
            params["labelSelector"] = label_selector
        r = self._req("GET", self._path(kind, namespace), params=params,
                      stream=True, timeout=timeout_s)
        try:
            for line in r.iter_lines():
                if not line:
                    continue
                ev = json.loads(line)
                typ = ev.get("type", "")
                obj = ev.get("object", {})
                if typ == "ERROR":
                    code = (obj.get("code")
                            or obj.get("status", {}).get("code") or 500)
                    raise ApiError(int(code), "watch error", json.dumps(obj))
                yield typ, obj
        finally:
            r.close()

    # ----------------------------------------------------- leader lease

    def acquire_lease(self, name: str, namespace: str, identity: str,
                      duration_s: int = 15) -> bool:
        """coordination.k8s.io Lease-based leader election (one round)."""

        import datetime
        now = datetime.datetime.utcnow().strftime("%Y-%m-%dT%H:%M:%S.%fZ")
        body = {
            "apiVersion": "coordination.k8s.io/v1", "kind": "Lease",
            "metadata": {"name": name, "namespace": namespace},
            "spec": {"holderIdentity": identity,
                     "leaseDurationSeconds": duration_s,
                     "renewTime": now},
        }
        cur = self.try_get("Lease", name, namespace)
        if cur is None:
            try:
                self.create(body, namespace)
                return True
            except ApiError as e:
                if e.conflict:
                    return False
                raise
        holder = cur.get("spec", {}).get("holderIdentity", "")
        renew = cur.get("spec", {}).get("renewTime", "")
        expired = False
        if renew:
            import datetime as dt
            try:
                t = dt.datetime.strptime(renew[:19], "%Y-%m-%dT%H:%M:%S")
                expired = (dt.datetime.utcnow() - t).total_seconds() > \
                    cur["spec"].get("leaseDurationSeconds", duration_s) * 2
            except ValueError:
                expired = True
        if holder == identity or expired or not holder:
            body["metadata"]["resourceVersion"] = \
                cur["metadata"].get("resourceVersion", "")
            try:
                self.update(body)
                return True
            except ApiError as e:
                if e.conflict:
                    return False
                raise
        return False
