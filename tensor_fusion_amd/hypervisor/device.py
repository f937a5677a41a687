"""ctypes binding to libaccelerator_amd.so + the device controller.

Reference: pkg/hypervisor/device/accelerator.go (purego FFI) +
controller.go (periodic discovery, metrics/process passthrough, partition
split). Our ABI is native/accelerator/tf_accelerator.h.
"""
from __future__ import annotations

import ctypes
import os
import threading
import time
from dataclasses import dataclass, field
from typing import Callable, Dict, List, Optional

from .. import constants as C

UUID_LEN = 64
NAME_LEN = 96


class _CDevice(ctypes.Structure):
    _fields_ = [
        ("uuid", ctypes.c_char * UUID_LEN),
        ("name", ctypes.c_char * NAME_LEN),
        ("index", ctypes.c_int32),
        ("numa_node", ctypes.c_int32),
        ("vram_total_bytes", ctypes.c_uint64),
        ("compute_units", ctypes.c_uint32),
        ("xcd_count", ctypes.c_uint32),
        ("bdf", ctypes.c_uint64),
        ("fp16_tflops", ctypes.c_double),
        ("is_mock", ctypes.c_uint32),
    ]


class _CMetrics(ctypes.Structure):
    _fields_ = [
        ("gfx_activity_percent", ctypes.c_uint32),
        ("umc_activity_percent", ctypes.c_uint32),
        ("vram_used_bytes", ctypes.c_uint64),
        ("vram_total_bytes", ctypes.c_uint64),
        ("power_w", ctypes.c_uint32),
        ("temp_c", ctypes.c_uint32),
        ("clock_mhz", ctypes.c_uint32),
    ]


class _CProc(ctypes.Structure):
    _fields_ = [
        ("pid", ctypes.c_int32),
        ("vram_bytes", ctypes.c_uint64),
        ("gfx_busy_ns", ctypes.c_uint64),
        ("cu_occupancy", ctypes.c_uint32),
        ("name", ctypes.c_char * NAME_LEN),
    ]


@dataclass
class DeviceInfo:
    uuid: str
    name: str
    index: int
    numa_node: int
    vram_total: int
    compute_units: int
    xcd_count: int
    fp16_tflops: float
    is_mock: bool


@dataclass
class DeviceMetrics:
    gfx_activity: int = 0
    umc_activity: int = 0
    vram_used: int = 0
    vram_total: int = 0


@dataclass
class ProcInfo:
    pid: int
    vram_bytes: int
    gfx_busy_ns: int
    cu_occupancy: int
    name: str


class Accelerator:
    """Thin safe wrapper over the C ABI."""

    def __init__(self, lib_path: Optional[str] = None, mock_devices: int = 0):
        if lib_path is None:
            lib_path = os.path.join(
                os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
                "_native", C.AcceleratorLibName)
        prev_mock = os.environ.get("TF_ACCEL_MOCK")
        if mock_devices:
            os.environ["TF_ACCEL_MOCK"] = str(mock_devices)
        self._lib = ctypes.CDLL(lib_path)
        self._lib.tf_accel_init.restype = ctypes.c_int
        # the C state is process-global; a changed mock count (tests) needs
        # a shutdown→init cycle so the env is re-read
        if mock_devices and prev_mock != str(mock_devices):
            self._lib.tf_accel_shutdown()
        rc = self._lib.tf_accel_init()
        if rc != 0:
            raise RuntimeError(f"tf_accel_init failed: {rc}")

    def shutdown(self):
        self._lib.tf_accel_shutdown()

    def devices(self) -> List[DeviceInfo]:
        arr = (_CDevice * 32)()
        n = ctypes.c_int()
        rc = self._lib.tf_accel_get_devices(arr, 32, ctypes.byref(n))
        if rc != 0:
            raise RuntimeError(f"tf_accel_get_devices: {rc}")
        out = []
        for i in range(n.value):
            d = arr[i]
            out.append(DeviceInfo(
                uuid=d.uuid.decode(), name=d.name.decode(), index=d.index,
                numa_node=d.numa_node, vram_total=d.vram_total_bytes,
                compute_units=d.compute_units, xcd_count=d.xcd_count,
                fp16_tflops=d.fp16_tflops, is_mock=bool(d.is_mock)))
        return out

    # ---- AMD compute/memory partition modes (SPX..CPX x NPS1..NPS8) ----

    NOT_SUPPORTED = 2  # TF_ACCEL_NOT_SUPPORTED

    def compute_partition(self, device: int) -> Optional[str]:
        """Current device-global compute-partition mode, or None when the
        amd-smi build lacks the API."""

        buf = ctypes.create_string_buffer(32)
        rc = self._lib.tf_accel_get_compute_partition(device, buf, 32)
        if rc == self.NOT_SUPPORTED:
            return None
        if rc != 0:
            raise RuntimeError(f"tf_accel_get_compute_partition: {rc}")
        return buf.value.decode()

    def set_compute_partition(self, device: int, mode: str) -> bool:
        rc = self._lib.tf_accel_set_compute_partition(device, mode.encode())
        if rc == self.NOT_SUPPORTED:
            return False
        if rc != 0:
            raise RuntimeError(
                f"tf_accel_set_compute_partition({mode}): {rc}")
        return True

    def memory_partition(self, device: int) -> Optional[str]:
        buf = ctypes.create_string_buffer(32)
        rc = self._lib.tf_accel_get_memory_partition(device, buf, 32)
        if rc == self.NOT_SUPPORTED:
            return None
        if rc != 0:
            raise RuntimeError(f"tf_accel_get_memory_partition: {rc}")
        return buf.value.decode()

    def set_memory_partition(self, device: int, mode: str) -> bool:
        rc = self._lib.tf_accel_set_memory_partition(device, mode.encode())
        if rc == self.NOT_SUPPORTED:
            return False
        if rc != 0:
            raise RuntimeError(f"tf_accel_set_memory_partition({mode}): {rc}")
        return True

    def topology(self, n: int) -> List[List[int]]:
        arr = (ctypes.c_int32 * (n * n))()
        rc = self._lib.tf_accel_get_topology(arr, n)
        if rc != 0:
            raise RuntimeError(f"tf_accel_get_topology: {rc}")
        return [[arr[i * n + j] for j in range(n)] for i in range(n)]

    def metrics(self, device: int) -> DeviceMetrics:
        m = _CMetrics()
        rc = self._lib.tf_accel_get_metrics(device, ctypes.byref(m))
        if rc != 0:
            raise RuntimeError(f"tf_accel_get_metrics({device}): {rc}")
        return DeviceMetrics(m.gfx_activity_percent, m.umc_activity_percent,
                             m.vram_used_bytes, m.vram_total_bytes)

    def processes(self, device: int, max_procs: int = 128) -> List[ProcInfo]:
        arr = (_CProc * max_procs)()
        n = ctypes.c_int()
        rc = self._lib.tf_accel_get_processes(device, arr, max_procs,
                                              ctypes.byref(n))
        if rc != 0:
            return []
        return [ProcInfo(arr[i].pid, arr[i].vram_bytes, arr[i].gfx_busy_ns,
                         arr[i].cu_occupancy, arr[i].name.decode())
                for i in range(n.value)]

    def cu_mask_env_for_percent(self, device: int, percent: float) -> str:
        buf = ctypes.create_string_buffer(256)
        rc = self._lib.tf_accel_compose_percent_mask_env(
            device, ctypes.c_double(percent), buf, 256)
        if rc != 0:
            raise RuntimeError("compose_percent_mask_env failed")
        return buf.value.decode()

    def cu_mask_env_for_xcds(self, device: int, xcds: List[int]) -> str:
        arr = (ctypes.c_int32 * len(xcds))(*xcds)
        buf = ctypes.create_string_buffer(256)
        rc = self._lib.tf_accel_compose_cu_mask_env(device, arr, len(xcds),
                                                    buf, 256)
        if rc != 0:
            raise RuntimeError("compose_cu_mask_env failed")
        return buf.value.decode()

    def assign_partition(self, device: int, xcds: List[int]) -> bool:
        arr = (ctypes.c_int32 * len(xcds))(*xcds)
        return self._lib.tf_accel_assign_partition(device, arr, len(xcds)) == 0

    def remove_partition(self, device: int, xcds: List[int]) -> bool:
        arr = (ctypes.c_int32 * len(xcds))(*xcds)
        return self._lib.tf_accel_remove_partition(
            device, arr, len(xcds)) == 0

    def snapshot(self, pid: int, dest: str) -> int:
        return self._lib.tf_accel_snapshot(pid, dest.encode())

    def resume(self, pid: int, src: str) -> int:
        return self._lib.tf_accel_resume(pid, src.encode())

    def device_count(self) -> int:
        n = ctypes.c_int(0)
        if self._lib.tf_accel_device_count(ctypes.byref(n)) != 0:
            raise RuntimeError("tf_accel_device_count failed")
        return n.value

    # The C side stores the RAW function pointer: thunks must outlive
    # every library that might call them, so they are pinned process-
    # wide (a dropped thunk after GC is a guaranteed segfault on the
    # next log emission — found by test-order shuffling).
    _LOG_THUNKS: list = []
    _LOG_CB_T = ctypes.CFUNCTYPE(None, ctypes.c_int, ctypes.c_char_p)

    def register_log_callback(self, fn) -> None:
        """fn(level:int, msg:str)."""

        def thunk(level, msg):
            try:
                fn(int(level), (msg or b"").decode(errors="replace"))
            except Exception:
                pass
        cb = self._LOG_CB_T(thunk)
        Accelerator._LOG_THUNKS.append(cb)
        self._lib.tf_accel_register_log_callback(cb)

    def unregister_log_callback(self) -> None:
        self._lib.tf_accel_register_log_callback(
            ctypes.cast(None, self._LOG_CB_T))


class DeviceController:
    """Discovery + periodic refresh + change handlers (reference
    device/controller.go:1-465)."""

    def __init__(self, accel: Accelerator, rediscover_interval_s: float = 3600):
        self.accel = accel
        self.interval = rediscover_interval_s
        self._devices: List[DeviceInfo] = []
        self._handlers: List[Callable[[List[DeviceInfo]], None]] = []
        self._mu = threading.Lock()
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None
        self.discover()

    def discover(self) -> List[DeviceInfo]:
        devs = self.accel.devices()
        with self._mu:
            changed = [d.uuid for d in devs] != [d.uuid for d in self._devices]
            self._devices = devs
        if changed:
            for h in list(self._handlers):
                h(devs)
        return devs

    def devices(self) -> List[DeviceInfo]:
        with self._mu:
            return list(self._devices)

    def device_by_uuid(self, uuid: str) -> Optional[DeviceInfo]:
        for d in self.devices():
            if d.uuid == uuid:
                return d
        return None

    def on_devices_changed(self, h: Callable[[List[DeviceInfo]], None]):
        self._handlers.append(h)
        h(self.devices())

    def metrics(self, index: int) -> DeviceMetrics:
        return self.accel.metrics(index)

    def processes(self, index: int) -> List[ProcInfo]:
        return self.accel.processes(index)

    def start(self):
        def loop():
            while not self._stop.wait(self.interval):
                try:
                    self.discover()
                except Exception:
                    pass
        self._thread = threading.Thread(target=loop, daemon=True)
        self._thread.start()

    def stop(self):
        self._stop.set()
