"""kubelet device plugin advertising tensor-fusion.ai/index-N.

The scheduler decides GPU placement (PreBind annotations); kubelet only
needs a per-pod hook to hand the worker container its environment and
/dev nodes. Like the reference (deviceplugin.go:48-366), the hypervisor
registers one small device plugin per index resource name
`tensor-fusion.ai/index-N` — the webhook puts exactly one such extended
resource on each worker pod (mutator.py:253), kubelet calls Allocate on
the matching plugin, and the plugin resolves the pod bound to that index
on this node to compose env (HIP_VISIBLE_DEVICES, TF_SHM_PATH, limiter
vars) + device nodes (/dev/kfd, /dev/dri/renderD*).

Implementation is pure grpcio over runtime-built protobuf descriptors
(dp_proto.py) — wire-compatible with a real kubelet's
v1beta1 Registration/DevicePlugin services over unix sockets.
"""
from __future__ import annotations

import os
import threading
from concurrent.futures import ThreadPoolExecutor
from queue import Empty, Queue
from typing import Callable, Dict, List, Optional

import grpc

from .. import constants as C
from .dp_proto import (API_VERSION, KUBELET_SOCKET, M_ALLOCATE,
                       M_LISTWATCH, M_OPTIONS, M_PREFERRED, M_PRESTART,
                       MSG, REGISTRATION_SERVICE)

# Allocate() answer for one pod index: env + device nodes + annotations.
AllocationResolver = Callable[[int], Optional[dict]]
# dict: {"env": {..}, "devices": [{"host_path","container_path",
#        "permissions"}], "annotations": {..}, "mounts": [...]}


def default_device_nodes() -> List[dict]:
    """The /dev set every ROCm workload container needs."""

    nodes = [{"host_path": "/dev/kfd", "container_path": "/dev/kfd",
              "permissions": "rw"}]
    dri = "/dev/dri"
    if os.path.isdir(dri):
        for n in sorted(os.listdir(dri)):
            if n.startswith("renderD"):
                nodes.append({"host_path": f"{dri}/{n}",
                              "container_path": f"{dri}/{n}",
                              "permissions": "rw"})
    return nodes


class _PluginServicer(grpc.GenericRpcHandler):
    """One DevicePlugin service instance for one index resource."""

    def __init__(self, index: int, resolver: AllocationResolver,
                 slots: int = 8):
        self.index = index
        self.resolver = resolver
        self.slots = slots
        self._watch_queues: List[Queue] = []
        self._stopped = threading.Event()

    # -------------------------------------------------- grpc dispatch

    def service(self, handler_call_details):
        method = handler_call_details.method
        if method == M_OPTIONS:
            return grpc.unary_unary_rpc_method_handler(
                self._options,
                request_deserializer=MSG["Empty"].FromString,
                response_serializer=lambda m: m.SerializeToString())
        if method == M_LISTWATCH:
            return grpc.unary_stream_rpc_method_handler(
                self._list_and_watch,
                request_deserializer=MSG["Empty"].FromString,
                response_serializer=lambda m: m.SerializeToString())
        if method == M_ALLOCATE:
            return grpc.unary_unary_rpc_method_handler(
                self._allocate,
                request_deserializer=MSG["AllocateRequest"].FromString,
                response_serializer=lambda m: m.SerializeToString())
        if method == M_PRESTART:
            return grpc.unary_unary_rpc_method_handler(
                lambda req, ctx: MSG["PreStartContainerResponse"](),
                request_deserializer=MSG["PreStartContainerRequest"].FromString,
                response_serializer=lambda m: m.SerializeToString())
        if method == M_PREFERRED:
            return grpc.unary_unary_rpc_method_handler(
                lambda req, ctx: MSG["PreferredAllocationResponse"](),
                request_deserializer=MSG["PreferredAllocationRequest"].FromString,
                response_serializer=lambda m: m.SerializeToString())
        return None

    # ---------------------------------------------------- handlers

    def _options(self, request, context):
        return MSG["DevicePluginOptions"](pre_start_required=False)

    def _devices(self):
        return [MSG["Device"](ID=f"{self.index}-{i}", health="Healthy")
                for i in range(1, self.slots + 1)]

    def _list_and_watch(self, request, context):
        yield MSG["ListAndWatchResponse"](devices=self._devices())
        q: Queue = Queue()
        self._watch_queues.append(q)
        try:
            while not self._stopped.is_set():
                try:
                    q.get(timeout=1.0)
                except Empty:
                    continue
                yield MSG["ListAndWatchResponse"](devices=self._devices())
        finally:
            try:
                self._watch_queues.remove(q)
            except ValueError:
                pass

    def refresh(self):
        for q in list(self._watch_queues):
            q.put(1)

    def _allocate(self, request, context):
        out = MSG["AllocateResponse"]()
        for _creq in request.container_requests:
            cresp = out.container_responses.add()
            info = self.resolver(self.index)
            if info is None:
                # worker not registered yet: still return the device
                # nodes so the container can start; env follows via the
                # hypervisor HTTP contract (/api/v1/pod)
                info = {"env": {}, "devices": default_device_nodes(),
                        "annotations": {}, "mounts": []}
            for k, v in (info.get("env") or {}).items():
                cresp.envs[k] = str(v)
            for d in info.get("devices") or []:
                cresp.devices.add(container_path=d["container_path"],
                                  host_path=d["host_path"],
                                  permissions=d.get("permissions", "rw"))
            for m in info.get("mounts") or []:
                cresp.mounts.add(container_path=m["container_path"],
                                 host_path=m["host_path"],
                                 read_only=bool(m.get("read_only")))
            for k, v in (info.get("annotations") or {}).items():
                cresp.annotations[k] = str(v)
        return out

    def stop(self):
        self._stopped.set()
        self.refresh()


class DevicePluginManager:
    """Registers index-1..index-N plugins with kubelet and serves them.

    `resolver(index)` is supplied by the hypervisor's allocation
    controller: it maps a pod index on this node to the composed worker
    environment (allocation.py AllocateWorkerDevices output).
    """

    def __init__(self, resolver: AllocationResolver,
                 socket_dir: str = "/var/lib/kubelet/device-plugins",
                 kubelet_socket: str = KUBELET_SOCKET,
                 max_indexes: int = C.MaxWorkersPerNode):
        self.resolver = resolver
        self.socket_dir = socket_dir
        self.kubelet_socket = kubelet_socket
        self.max_indexes = max_indexes
        self._servers: List[grpc.Server] = []
        self._plugins: Dict[int, _PluginServicer] = {}
        self._stop = threading.Event()
        self._watcher: Optional[threading.Thread] = None
        self.registrations = 0  # total register RPCs (visible to tests)

    # ------------------------------------------------------------ serve

    def start(self, register: bool = True) -> "DevicePluginManager":
        os.makedirs(self.socket_dir, exist_ok=True)
        for idx in range(1, self.max_indexes + 1):
            plugin = _PluginServicer(idx, self.resolver)
            endpoint = f"tf-index-{idx}.sock"
            path = os.path.join(self.socket_dir, endpoint)
            try:
                os.unlink(path)
            except OSError:
                pass
            server = grpc.server(
                ThreadPoolExecutor(max_workers=2),
                options=[("grpc.max_receive_message_length", 1 << 20)])
            server.add_generic_rpc_handlers((plugin,))
            server.add_insecure_port(f"unix://{path}")
            server.start()
            self._servers.append(server)
            self._plugins[idx] = plugin
            if register:
                self._register(idx, endpoint)
        if register:
            self._watcher = threading.Thread(
                target=self._watch_kubelet, daemon=True,
                name="dp-kubelet-watch")
            self._watcher.start()
        return self

    def _kubelet_ino(self):
        """Identity of the current kubelet socket: (inode, ctime_ns).
        Inode alone is unreliable — tmpfs reuses inode numbers
        immediately, so a fast restart could look unchanged."""

        try:
            st = os.stat(self.kubelet_socket)
            return (st.st_ino, st.st_ctime_ns)
        except OSError:
            return None

    def _watch_kubelet(self, poll_s: float = 2.0):
        """kubelet restarts wipe its plugin registry and recreate its
        registration socket — detect the new socket inode and
        re-register every index plugin (reference deviceplugin.go
        restart loop)."""

        last = self._kubelet_ino()
        while not self._stop.wait(getattr(self, "_watch_kubelet_poll",
                                          poll_s)):
            ident = self._kubelet_ino()
            if ident is not None and ident != last:
                for idx in self._plugins:
                    try:
                        self._register(idx, f"tf-index-{idx}.sock")
                    except grpc.RpcError:
                        ident = None  # kubelet not ready: retry next poll
                        break
            if ident is not None:
                last = ident

    def _register(self, idx: int, endpoint: str, timeout: float = 5.0):
        ch = grpc.insecure_channel(f"unix://{self.kubelet_socket}")
        try:
            stub = ch.unary_unary(
                REGISTRATION_SERVICE,
                request_serializer=lambda m: m.SerializeToString(),
                response_deserializer=MSG["Empty"].FromString)
            req = MSG["RegisterRequest"](
                version=API_VERSION, endpoint=endpoint,
                resource_name=f"{C.IndexResourcePrefix}{idx}")
            stub(req, timeout=timeout)
            self.registrations += 1
        finally:
            ch.close()

    def refresh(self):
        for p in self._plugins.values():
            p.refresh()

    def stop(self, grace: float = 0.5):
        self._stop.set()
        for p in self._plugins.values():
            p.stop()
        for s in self._servers:
            s.stop(grace)
