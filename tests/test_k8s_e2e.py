"""BASELINE config 1 — operator + gpuallocator on a (fake) cluster with
8 fake-GPU nodes, end to end over the Kubernetes wire:

    kubectl-equivalent client ──HTTP──▶ apiserver
        pod CREATE ──AdmissionReview HTTP──▶ operator webhook (mutates)
        watch streams ──▶ K8sStore informers ──▶ controllers/scheduler
        scheduler binds via the pods/binding subresource
        connection URL lands in the TensorFusionConnection CR status

Everything the operator does here goes through REST + watch — no direct
store access from the "user" side. (kind itself cannot run in this
offline environment; the fake apiserver speaks the same wire, see
tensor_fusion_amd/k8s/fake_apiserver.py.)

Reference: cmd/main.go:131-297, webhook pod_webhook.go:84, scheduler
PreBind gpuresources.go:882, connection controller :136.
"""
import socket
import threading
import time

import pytest
import uvicorn

import tensor_fusion_amd.constants as C
from tensor_fusion_amd.api import types as T
from tensor_fusion_amd.k8s import serde
from tensor_fusion_amd.k8s.bridge import K8sStore
from tensor_fusion_amd.k8s.client import K8sClient
from tensor_fusion_amd.k8s.crdgen import all_crds
from tensor_fusion_amd.k8s.fake_apiserver import serve_in_thread
from tensor_fusion_amd.operator import build_operator
from tensor_fusion_amd.server.webhook_server import create_webhook_app

NODES = 8
GPUS_PER_NODE = 8


def _free_port() -> int:
    with socket.socket() as sk:
        sk.bind(("127.0.0.1", 0))
        return sk.getsockname()[1]


@pytest.fixture(scope="module")
def cluster():
    """Fake apiserver + CRDs + 8 fake-GPU nodes + kube-backed operator
    with its AdmissionReview webhook registered and served over HTTP."""

    srv, base, us = serve_in_thread()
    kubectl = K8sClient(base)
    for crd in all_crds().values():
        kubectl.create(crd)

    # the "cluster": 8 nodes × 8 MI355X GPUs, published like the node
    # plane would (Node + GPU CRs through the API)
    pool = T.GPUPool()
    pool.meta.name = "pool-a"
    kubectl.create(serde.to_k8s(pool))
    for n in range(NODES):
        node = T.Node()
        node.meta.name = f"node-{n}"
        kubectl.create(serde.to_k8s(node))
        for i in range(GPUS_PER_NODE):
            g = T.GPU()
            g.meta.name = f"node-{n}-g{i}"
            g.status.uuid = f"uuid-{g.meta.name}"
            g.status.node = f"node-{n}"
            g.status.pool = "pool-a"
            g.status.capacity = T.Resource(C.MI355X_BF16_TFLOPS,
                                           C.MI355X_VRAM_BYTES, 100.0)
            g.status.available = T.Resource(C.MI355X_BF16_TFLOPS,
                                            C.MI355X_VRAM_BYTES, 100.0)
            wire = serde.to_k8s(g)
            created = kubectl.create(wire)
            created["status"] = wire["status"]
            kubectl.update_status(created)

    # kube-backed operator
    store = K8sStore(K8sClient(base), namespace="default").start()
    op = build_operator(store=store)

    # serve the AdmissionReview webhook and register it (url clientConfig,
    # like a kind setup with host networking)
    whport = _free_port()
    whapp = create_webhook_app(op.mutator)
    whsrv = uvicorn.Server(uvicorn.Config(whapp, host="127.0.0.1",
                                          port=whport, log_level="error"))
    threading.Thread(target=whsrv.run, daemon=True).start()
    import requests
    for _ in range(100):
        try:
            if requests.get(f"http://127.0.0.1:{whport}/healthz",
                            timeout=1).ok:
                break
        except Exception:
            time.sleep(0.05)
    kubectl.create({
        "apiVersion": "admissionregistration.k8s.io/v1",
        "kind": "MutatingWebhookConfiguration",
        "metadata": {"name": "tensor-fusion-mutating-webhook"},
        "webhooks": [{
            "name": "mpod.tensor-fusion.ai",
            "clientConfig": {
                "url": f"http://127.0.0.1:{whport}/mutate-v1-pod"},
            "objectSelector": {
                "matchLabels": {C.LabelEnabled: "true"}},
        }],
    })

    yield kubectl, op
    store.stop()
    whsrv.should_exit = True
    us.should_exit = True


def _wait(fn, timeout=20.0, op=None):
    deadline = time.time() + timeout
    while time.time() < deadline:
        if op is not None:
            op.tick()
        r = fn()
        if r:
            return r
        time.sleep(0.05)
    return fn()


class TestKindStyleE2E:
    def test_fractional_vgpu_pod_schedules_over_the_wire(self, cluster):
        kubectl, op = cluster
        # the user's pod: only labels + annotations, created via the API
        pod = {
            "apiVersion": "v1", "kind": "Pod",
            "metadata": {
                "name": "app-1", "namespace": "default",
                "labels": {C.LabelEnabled: "true"},
                "annotations": {
                    C.AnnoTflopsRequest: "600",
                    C.AnnoVramRequest: str(48 << 30),
                },
            },
            "spec": {"containers": [{"name": "main", "image": "app:1"}]},
        }
        kubectl.create(pod)

        # webhook ran during admission: the stored pod is already mutated
        # (client env injected; worker pods get the custom schedulerName)
        stored = kubectl.get("Pod", "app-1", "default")
        env = {e["name"]: e.get("value") for e in
               stored["spec"]["containers"][0].get("env", [])}
        assert C.EnvConnectionName in env
        assert env[C.EnvConnectionName] == "app-1-conn"

        # operator loops: workload → worker pod → schedule → bind
        worker = _wait(lambda: next(
            (p for p in kubectl.list_items(
                "Pod", "default",
                label_selector=f"{C.LabelComponent}={C.ComponentWorker}")
             if p.get("spec", {}).get("nodeName")), None), op=op)
        assert worker, "worker pod should be created and bound on the wire"
        assert worker["spec"]["nodeName"].startswith("node-")
        annos = worker["metadata"]["annotations"]
        assert C.AnnoGpuIds in annos and annos[C.AnnoGpuIds]

        # connection URL published in CR status on the wire
        conn = _wait(lambda: (kubectl.try_get(
            "TensorFusionConnection", "app-1-conn", "default") or {}
        ).get("status", {}).get("connectionUrl"), op=op)
        assert conn and conn.startswith("native+")

        # allocator accounted the device on the wire (GPU CR status sync)
        def allocated():
            for i in range(GPUS_PER_NODE):
                g = kubectl.get("GPU", f"{worker['spec']['nodeName']}-g{i}")
                avail = g.get("status", {}).get("available", {})
                if avail.get("tflops", C.MI355X_BF16_TFLOPS) <= \
                        C.MI355X_BF16_TFLOPS - 600:
                    return g
            return None
        g = _wait(allocated, op=op)
        assert g is not None, "GPU CR status should show the allocation"

    def test_workload_scale_down_releases_on_the_wire(self, cluster):
        kubectl, op = cluster
        wl = _wait(lambda: kubectl.try_get(
            "TensorFusionWorkload", "app-1-wl", "default"), op=op)
        assert wl
        kubectl.patch("TensorFusionWorkload", "app-1-wl",
                      {"spec": {"replicas": 0}}, namespace="default")
        gone = _wait(lambda: not [
            p for p in kubectl.list_items(
                "Pod", "default",
                label_selector=f"{C.LabelComponent}={C.ComponentWorker}")
            if p["metadata"].get("labels", {}).get(C.LabelWorkload)
            == "app-1-wl"], op=op)
        assert gone, "workers should be deleted after scale to 0"

    def test_eight_nodes_visible_to_allocator(self, cluster):
        kubectl, op = cluster
        assert len(op.allocator.gpus()) == NODES * GPUS_PER_NODE
        nodes = {g.status.node for g in op.allocator.gpus()}
        assert len(nodes) == NODES


def test_operator_cli_k8s_mode(tmp_path):
    """`python -m tensor_fusion_amd.operator --k8s --install-crds` as a
    real process against the (fake) apiserver via TF_K8S_URL: CRDs are
    applied, the webhook server comes up, informers sync, and the
    operator HTTP endpoint serves — the deployment entrypoint
    (deploy/manifests/operator.yaml) exercised end to end."""

    import os
    import subprocess
    import sys

    import requests
    srv, base, us = serve_in_thread()
    http_port, wh_port = _free_port(), _free_port()
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    env = dict(os.environ, TF_K8S_URL=base)
    proc = subprocess.Popen(
        [sys.executable, "-m", "tensor_fusion_amd.operator", "--k8s",
         "--install-crds", "--http-port", str(http_port),
         "--webhook-port", str(wh_port),
         "--leader-elect", "--webhook-workers", "2"],
        env=env, cwd=repo, stdout=subprocess.PIPE,
        stderr=subprocess.PIPE, text=True)
    try:
        kubectl = K8sClient(base)
        deadline = time.time() + 60
        up = False
        while time.time() < deadline:
            if proc.poll() is not None:
                break
            try:
                if requests.get(
                        f"http://127.0.0.1:{wh_port}/healthz",
                        timeout=1).ok:
                    up = True
                    break
            except Exception:
                time.sleep(0.2)
        assert up, (proc.poll(), proc.stderr.read(4000)
                    if proc.poll() is not None else "webhook not up")
        # CRDs were applied by --install-crds
        crds = kubectl.list_items("CustomResourceDefinition")
        assert len(crds) == 12, [c["metadata"]["name"] for c in crds]
        # a GPU created through the API reaches the operator's informers
        g = T.GPU()
        g.meta.name = "cli-g0"
        g.status.node = "cli-n0"
        kubectl.create(serde.to_k8s(g))
        # webhook admission over the wire mutates a pod
        r = requests.post(
            f"http://127.0.0.1:{wh_port}/mutate-v1-pod",
            json={"apiVersion": "admission.k8s.io/v1",
                  "kind": "AdmissionReview",
                  "request": {"uid": "u1", "object": {
                      "metadata": {"name": "p1", "namespace": "default",
                                   "labels": {C.LabelEnabled: "true"},
                                   "annotations": {
                                       C.AnnoTflopsRequest: "100"}},
                      "spec": {"containers": [{"name": "main"}]}}}},
            timeout=10)
        assert r.ok and r.json()["response"]["allowed"] is True
        assert "patch" in r.json()["response"]
    finally:
        proc.terminate()
        try:
            proc.wait(timeout=10)
        except subprocess.TimeoutExpired:
            proc.kill()
        us.should_exit = True


def test_node_plane_kubernetes_backend_over_the_wire(tmp_path):
    """Node onboarding over the wire: a hypervisor built
    with --backend kubernetes semantics (K8sStore) discovers its (mock)
    devices and publishes GPU + GPUNode CRs THROUGH the apiserver; a
    worker pod bound to the node appears via the watch and gets its shm
    page + allocation; deleting the pod tears the worker down."""

    import tensor_fusion_amd.constants as C
    from tensor_fusion_amd.hypervisor.main import build_hypervisor

    srv, base, us = serve_in_thread()
    kubectl = K8sClient(base)
    for crd in all_crds().values():
        kubectl.create(crd)
    store = K8sStore(K8sClient(base), namespace="default").start()
    try:
        devices, workers, erl, backend = build_hypervisor(
            node="knode-0", mock_devices=2,
            shm_root=str(tmp_path / "shm"), store=store)
        devices.start()
        backend.start()

        # GPU CRs (with capacity + topology) and the GPUNode exist on
        # the WIRE, not just locally
        gpus = _wait(lambda: kubectl.list_items("GPU") or None)
        assert gpus and len(gpus) == 2
        st = gpus[0]["status"]
        assert st["capacity"]["vram"] == C.MI355X_VRAM_BYTES
        assert st["node"] == "knode-0"
        gn = kubectl.get("GPUNode", "knode-0")
        assert gn["status"]["gpuCount"] == 2

        # a worker pod bound to this node (as the scheduler would leave
        # it) flows in via the informer → worker + shm appear
        uuid0 = gpus[0]["status"]["uuid"]
        pod = {
            "apiVersion": "v1", "kind": "Pod",
            "metadata": {
                "name": "w-kube", "namespace": "default",
                "labels": {C.LabelComponent: C.ComponentWorker},
                "annotations": {
                    C.AnnoGpuIds: uuid0,
                    C.AnnoContainerGpus: gpus[0]["metadata"]["name"],
                    C.AnnoVramRequest: str(8 << 30),
                    C.AnnoVramLimit: str(8 << 30),
                    C.AnnoTflopsLimit: "600",
                },
            },
            "spec": {"containers": [{"name": "main"}],
                     "nodeName": "knode-0"},
        }
        kubectl.create(pod)
        w = _wait(lambda: workers.get("default/w-kube"))
        assert w is not None
        assert w.allocation.spec.gpu_uuids == [uuid0]
        assert (tmp_path / "shm" / "default" / "w-kube" / "shm").exists()

        kubectl.delete("Pod", "w-kube", "default")
        gone = _wait(lambda: workers.get("default/w-kube") is None)
        assert gone
    finally:
        store.stop()
        us.should_exit = True


def test_leader_election_over_wire():
    """Two operator identities contend for the Lease over the real
    wire; the loser backs off, the winner renews, and once the holder
    stops renewing past the staleness window the loser takes over."""

    import time

    from tensor_fusion_amd.k8s.fake_apiserver import serve_in_thread
    from tensor_fusion_amd.operator import run_leader_election

    srv, url, us = serve_in_thread()
    c1 = K8sClient(url)
    c2 = K8sClient(url)

    lost = []
    id1 = run_leader_election(c1, "op-lease", "default", duration_s=1,
                              identity="op-1", on_lost=lambda: lost.append(1))
    assert id1 == "op-1"
    # second replica cannot take a held, renewed lease
    id2 = run_leader_election(c2, "op-lease", "default", duration_s=1,
                              identity="op-2", _max_wait=1.5)
    assert id2 == ""
    # holder stops renewing (simulate crash): acquire_lease from op-2
    # succeeds once the lease is 2x stale
    # break op-1's renewals (simulated crash) and wait out the
    # 2x-duration staleness window
    c1.base_url = "http://127.0.0.1:1"
    deadline = time.time() + 15
    got = False
    while time.time() < deadline:
        if c2.acquire_lease("op-lease", "default", "op-2", duration_s=1):
            got = True
            break
        time.sleep(0.3)
    assert got
    us.should_exit = True


def test_demo_cluster_script_runs(capsys):
    """tools/demo_cluster.py (the human-facing walkthrough) must keep
    working — it exercises the same wire flow this module asserts."""

    import importlib.util
    import pathlib
    spec = importlib.util.spec_from_file_location(
        "demo_cluster", pathlib.Path(__file__).resolve().parent.parent
        / "tools" / "demo_cluster.py")
    mod = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(mod)
    mod.main()
    out = capsys.readouterr().out
    assert "12 CRDs" in out
    assert "native+" in out, out  # connection URL published
    assert "demo complete." in out
