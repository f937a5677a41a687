"""kubelet device-plugin protocol tests (real gRPC over unix sockets).

A fake kubelet (its Registration service + a device-plugin client, the
two halves kubelet implements) talks to DevicePluginManager exactly like
kubelet does: Register → GetDevicePluginOptions → ListAndWatch →
Allocate. Wire format is the v1beta1 protobuf built in dp_proto.py.

Also covers the checkpoint coexistence detector
(kubelet_internal_checkpoint → GPU.used_by flip).
"""
import json
import threading
import time
from concurrent.futures import ThreadPoolExecutor

import grpc
import pytest

import tensor_fusion_amd.constants as C
from tensor_fusion_amd.api.store import Store
from tensor_fusion_amd.api.types import GPU, Resource
from tensor_fusion_amd.k8s.deviceplugin import (DevicePluginManager,
                                                default_device_nodes)
from tensor_fusion_amd.k8s.dp_proto import (API_VERSION, M_ALLOCATE,
                                            M_LISTWATCH, M_OPTIONS, MSG,
                                            REGISTRATION_SERVICE)
from tensor_fusion_amd.k8s.kubelet_checkpoint import (EXTERNAL_USED_BY,
                                                      CheckpointDetector)


class FakeKubeletRegistration(grpc.GenericRpcHandler):
    def __init__(self):
        self.registered = []
        self._ev = threading.Event()

    def service(self, details):
        if details.method == REGISTRATION_SERVICE:
            return grpc.unary_unary_rpc_method_handler(
                self._register,
                request_deserializer=MSG["RegisterRequest"].FromString,
                response_serializer=lambda m: m.SerializeToString())
        return None

    def _register(self, req, ctx):
        self.registered.append((req.resource_name, req.endpoint,
                                req.version))
        self._ev.set()
        return MSG["Empty"]()


@pytest.fixture()
def kubelet(tmp_path):
    sock = str(tmp_path / "kubelet.sock")
    servicer = FakeKubeletRegistration()
    server = grpc.server(ThreadPoolExecutor(max_workers=2))
    server.add_generic_rpc_handlers((servicer,))
    server.add_insecure_port(f"unix://{sock}")
    server.start()
    yield servicer, sock, str(tmp_path)
    server.stop(0.2)


def _resolver(index: int):
    if index == 2:
        return None  # unregistered worker
    return {
        "env": {"HIP_VISIBLE_DEVICES": "0",
                C.EnvShmPath: f"/run/tensor-fusion/shm/ns/pod-{index}/shm"},
        "devices": [{"host_path": "/dev/kfd", "container_path": "/dev/kfd",
                     "permissions": "rw"}],
        "annotations": {C.AnnoPodIndex: str(index)},
        "mounts": [{"host_path": "/run/tensor-fusion",
                    "container_path": "/run/tensor-fusion"}],
    }


class TestDevicePlugin:
    def test_register_listwatch_allocate(self, kubelet):
        servicer, ksock, sockdir = kubelet
        mgr = DevicePluginManager(_resolver, socket_dir=sockdir,
                                  kubelet_socket=ksock, max_indexes=3)
        mgr.start()
        try:
            deadline = time.time() + 10
            while len(servicer.registered) < 3 and time.time() < deadline:
                time.sleep(0.05)
            names = {r[0] for r in servicer.registered}
            assert names == {f"{C.IndexResourcePrefix}{i}"
                             for i in (1, 2, 3)}
            assert all(r[2] == API_VERSION for r in servicer.registered)

            # kubelet side: dial plugin 1's endpoint
            endpoint = next(e for (n, e, _) in servicer.registered
                            if n.endswith("-1"))
            ch = grpc.insecure_channel(f"unix://{sockdir}/{endpoint}")
            opts = ch.unary_unary(
                M_OPTIONS,
                request_serializer=lambda m: m.SerializeToString(),
                response_deserializer=MSG["DevicePluginOptions"].FromString
            )(MSG["Empty"](), timeout=5)
            assert opts.pre_start_required is False

            stream = ch.unary_stream(
                M_LISTWATCH,
                request_serializer=lambda m: m.SerializeToString(),
                response_deserializer=MSG["ListAndWatchResponse"].FromString
            )(MSG["Empty"](), timeout=5)
            first = next(iter(stream))
            assert len(first.devices) >= 1
            assert all(d.health == "Healthy" for d in first.devices)

            req = MSG["AllocateRequest"]()
            req.container_requests.add().devicesIDs.append("1-1")
            resp = ch.unary_unary(
                M_ALLOCATE,
                request_serializer=lambda m: m.SerializeToString(),
                response_deserializer=MSG["AllocateResponse"].FromString
            )(req, timeout=5)
            cr = resp.container_responses[0]
            assert cr.envs["HIP_VISIBLE_DEVICES"] == "0"
            assert cr.annotations[C.AnnoPodIndex] == "1"
            assert cr.devices[0].host_path == "/dev/kfd"
            assert cr.mounts[0].host_path == "/run/tensor-fusion"
            ch.close()
        finally:
            mgr.stop()

    def test_allocate_unresolved_returns_device_nodes(self, kubelet):
        servicer, ksock, sockdir = kubelet
        mgr = DevicePluginManager(_resolver, socket_dir=sockdir,
                                  kubelet_socket=ksock, max_indexes=2)
        mgr.start()
        try:
            endpoint = f"tf-index-2.sock"
            ch = grpc.insecure_channel(f"unix://{sockdir}/{endpoint}")
            req = MSG["AllocateRequest"]()
            req.container_requests.add().devicesIDs.append("2-1")
            resp = ch.unary_unary(
                M_ALLOCATE,
                request_serializer=lambda m: m.SerializeToString(),
                response_deserializer=MSG["AllocateResponse"].FromString
            )(req, timeout=5)
            cr = resp.container_responses[0]
            # no env (worker not yet known) but the /dev nodes are there
            assert len(cr.devices) >= 1
            assert cr.devices[0].host_path == "/dev/kfd"
        finally:
            mgr.stop()


def _mk_gpu(store, name, node, uuid, index):
    g = GPU()
    g.meta.name = name
    g.status.uuid = uuid
    g.status.index = index
    g.status.node = node
    g.status.capacity = Resource(2500.0, 288 << 30, 100.0)
    g.status.available = Resource(2500.0, 288 << 30, 100.0)
    store.create(g)
    return g


class TestCheckpointDetector:
    def test_foreign_devices_marked_and_restored(self, tmp_path):
        store = Store()
        _mk_gpu(store, "n0-g0", "n0", "uuid-aaa", 0)
        _mk_gpu(store, "n0-g1", "n0", "uuid-bbb", 1)
        _mk_gpu(store, "n1-g0", "n1", "uuid-ccc", 0)  # other node

        ckpt = tmp_path / "kubelet_internal_checkpoint"
        ckpt.write_text(json.dumps({
            "Data": {
                "RegisteredDevices": {"amd.com/gpu": ["uuid-aaa"]},
                "PodDeviceEntries": [{
                    "PodUID": "u1", "ContainerName": "c",
                    "ResourceName": "amd.com/gpu",
                    "DeviceIDs": {"0": ["uuid-aaa"]},
                }],
            },
            "Checksum": 1,
        }))
        det = CheckpointDetector(store, node="n0", path=str(ckpt))
        assert det.sync_once() == 1
        assert store.get("GPU", "n0-g0").status.used_by == EXTERNAL_USED_BY
        assert store.get("GPU", "n0-g1").status.used_by == "tensor-fusion"
        # other node untouched
        assert store.get("GPU", "n1-g0").status.used_by == "tensor-fusion"

        # plugin removed → device returns to the pool
        ckpt.write_text(json.dumps({"Data": {"RegisteredDevices": {},
                                             "PodDeviceEntries": []},
                                    "Checksum": 2}))
        assert det.sync_once() == 1
        assert store.get("GPU", "n0-g0").status.used_by == "tensor-fusion"

    def test_index_style_ids(self, tmp_path):
        store = Store()
        _mk_gpu(store, "n0-g0", "n0", "", 0)
        _mk_gpu(store, "n0-g1", "n0", "", 1)
        ckpt = tmp_path / "ck"
        ckpt.write_text(json.dumps({
            "Data": {"RegisteredDevices":
                     {"amd.com/gpu": ["/dev/dri/renderD1"]},
                     "PodDeviceEntries": []},
            "Checksum": 3}))
        det = CheckpointDetector(store, node="n0", path=str(ckpt))
        det.sync_once()
        assert store.get("GPU", "n0-g1").status.used_by == EXTERNAL_USED_BY
        assert store.get("GPU", "n0-g0").status.used_by == "tensor-fusion"

    def test_allocator_excludes_external_gpus(self, tmp_path):
        """The point of the detector: external GPUs never get allocated."""

        from tensor_fusion_amd.allocator.gpuallocator import GpuAllocator
        from tensor_fusion_amd.api.types import AllocRequest
        store = Store()
        _mk_gpu(store, "n0-g0", "n0", "uuid-aaa", 0)
        _mk_gpu(store, "n0-g1", "n0", "uuid-bbb", 1)
        ckpt = tmp_path / "ck"
        ckpt.write_text(json.dumps({
            "Data": {"RegisteredDevices": {"amd.com/gpu": ["uuid-aaa"]},
                     "PodDeviceEntries": []}, "Checksum": 4}))
        CheckpointDetector(store, node="n0", path=str(ckpt)).sync_once()
        alloc = GpuAllocator(store=store)
        req = AllocRequest(pod_name="p", workload="w",
                           request=Resource(100.0, 8 << 30, 10.0),
                           limit=Resource(100.0, 8 << 30, 10.0))
        scores, _ = alloc.check_quota_and_filter(req)
        assert "n0" in scores
        picked = alloc.pick_gpus(req, "n0")
        assert picked == ["n0-g1"]  # external GPU never allocated


class FakePodResourcesLister(grpc.GenericRpcHandler):
    """kubelet's PodResources service (the half kubelet implements)."""

    def __init__(self, pods):
        from tensor_fusion_amd.k8s.podresources import MSG as PR
        self.PR = PR
        self.pods = pods

    def service(self, details):
        from tensor_fusion_amd.k8s.podresources import M_LIST
        if details.method == M_LIST:
            return grpc.unary_unary_rpc_method_handler(
                self._list,
                request_deserializer=self.PR[
                    "ListPodResourcesRequest"].FromString,
                response_serializer=lambda m: m.SerializeToString())
        return None

    def _list(self, req, ctx):
        resp = self.PR["ListPodResourcesResponse"]()
        for ns, name, res, ids in self.pods:
            pr = resp.pod_resources.add(name=name, namespace=ns)
            c = pr.containers.add(name="main")
            d = c.devices.add(resource_name=res)
            d.device_ids.extend(ids)
        return resp


class TestPodResources:
    def test_list_and_maps(self, tmp_path):
        from tensor_fusion_amd.k8s.podresources import PodResourcesClient
        sock = str(tmp_path / "podres.sock")
        server = grpc.server(ThreadPoolExecutor(max_workers=2))
        server.add_generic_rpc_handlers((FakePodResourcesLister([
            ("default", "w0", "tensor-fusion.ai/index-3", ["3-1"]),
            ("default", "legacy", "amd.com/gpu", ["uuid-aaa"]),
            ("kube-system", "dns", "cpu", []),
        ]),))
        server.add_insecure_port(f"unix://{sock}")
        server.start()
        try:
            cli = PodResourcesClient(sock)
            pods = cli.list()
            assert len(pods) == 3
            # our index devices
            assert cli.device_map() == {"default/w0": ["3-1"]}
            # foreign GPU holders (live complement to the checkpoint file)
            assert cli.foreign_gpu_devices() == {
                "default/legacy": ["uuid-aaa"]}
        finally:
            server.stop(0.2)


class TestKubeletRestart:
    def test_reregister_after_kubelet_socket_recreated(self, tmp_path):
        """kubelet restart wipes its plugin registry and recreates its
        registration socket — the manager must notice the new socket
        inode and re-register every plugin (reference deviceplugin.go
        restart loop)."""

        sock = str(tmp_path / "kubelet.sock")

        def serve():
            servicer = FakeKubeletRegistration()
            server = grpc.server(ThreadPoolExecutor(max_workers=2))
            server.add_generic_rpc_handlers((servicer,))
            server.add_insecure_port(f"unix://{sock}")
            server.start()
            return servicer, server

        s1, srv1 = serve()
        mgr = DevicePluginManager(_resolver, socket_dir=str(tmp_path),
                                  kubelet_socket=sock, max_indexes=2)
        # fast poll for the test
        mgr._watch_kubelet_poll = 0.1
        mgr.start(register=True)
        try:
            assert len(s1.registered) == 2
            first = mgr.registrations
            # hand-drive the watcher logic across a "restart":
            old_ino = mgr._kubelet_ino()
            srv1.stop(0)
            import contextlib
            import os
            with contextlib.suppress(OSError):
                os.unlink(sock)  # grpc may remove its socket on stop
            s2, srv2 = serve()
            assert mgr._kubelet_ino() != old_ino
            deadline = time.time() + 10
            while time.time() < deadline and len(s2.registered) < 2:
                time.sleep(0.1)
            assert len(s2.registered) == 2, s2.registered
            assert mgr.registrations == first + 2
            srv2.stop(0.2)
        finally:
            mgr.stop()
