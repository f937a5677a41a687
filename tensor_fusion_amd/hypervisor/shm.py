"""Python mirror of the limiter shared-memory protocol.

Byte-layout contract: native/limiter/limiter_shm.h (asserted against the
compiled C++ side by tests/test_shm.py via tf_shm_layout_dump). The
hypervisor creates/updates these pages; the LD_PRELOAD limiter and the
remoting worker consume them on every HIP call.

Capability parity with reference soft_limiter_shm.go:140-1068 (V2 ERL
layout, open-not-truncate create, PID set, heartbeats).
"""
from __future__ import annotations

import mmap
import os
import struct
import time
from dataclasses import dataclass
from typing import List, Optional

SHM_SIZE = 4096
MAGIC = 0x5446414D
VERSION = 2

MAX_DEVICES = 16
MAX_PIDS = 64
UUID_LEN = 64

OFF_MAGIC = 0
OFF_VERSION = 4
OFF_DEV = 8
DEV_STRIDE = 160
OFF_DEVICE_COUNT = 2568
OFF_FLAGS = 2572
OFF_HEARTBEAT = 2576
OFF_HYP_HEARTBEAT = 2584
OFF_MUTEX = 2592
OFF_PID_COUNT = 2596
OFF_PIDS = 2600

# device-entry field offsets (relative to entry start)
E_UUID = 0
E_UP_LIMIT = 64
E_TOTAL_CUS = 68
E_MEM_LIMIT = 72
E_MEM_USED = 80
E_RATE = 88
E_CAPACITY = 96
E_TOKENS = 104
E_LAST_UPDATE = 112
E_ACTIVE = 120
E_LAUNCHES = 124
E_BLOCK_NS = 128
E_ALLOC_BYTES = 136
E_VMM_BYTES = 144  # remoting worker's VMM heap bytes (limiter_shm.h)

FLAG_FREEZE = 1 << 0
FLAG_VRAM_PRESSURE = 1 << 1

_u32 = struct.Struct("<I")
_u64 = struct.Struct("<Q")
_f64 = struct.Struct("<d")


@dataclass
class DeviceEntrySnapshot:
    uuid: str
    up_limit_percent: int
    total_cus: int
    mem_limit_bytes: int
    pod_memory_used: int
    erl_refill_rate: float
    erl_capacity: float
    erl_tokens: float
    erl_last_update_ns: int
    active: bool
    launch_count: int
    block_ns_total: int
    alloc_bytes_total: int
    vmm_bytes: int = 0  # remoting worker VMM heap (bypasses hipMalloc)

    @property
    def total_used(self) -> int:
        return self.pod_memory_used + self.vmm_bytes


class WorkerShm:
    """One worker's shm page, mapped read-write.

    `create()` uses open-not-truncate semantics: if a valid page already
    exists (worker survived a hypervisor restart) its counters are preserved
    (the reference fixed an O_TRUNC bug here — legacy.go:613-632).
    """

    def __init__(self, path: str, mm: mmap.mmap, fd: int):
        self.path = path
        self._mm = mm
        self._fd = fd

    # ------------------------------------------------------------ lifecycle

    @classmethod
    def create(cls, path: str) -> "WorkerShm":
        os.makedirs(os.path.dirname(path), exist_ok=True)
        fd = os.open(path, os.O_RDWR | os.O_CREAT, 0o666)
        st = os.fstat(fd)
        fresh = st.st_size < SHM_SIZE
        if fresh:
            os.ftruncate(fd, SHM_SIZE)
        mm = mmap.mmap(fd, SHM_SIZE)
        shm = cls(path, mm, fd)
        if fresh or shm.read_u32(OFF_MAGIC) != MAGIC:
            mm[:] = b"\x00" * SHM_SIZE
            shm.write_u32(OFF_MAGIC, MAGIC)
            shm.write_u32(OFF_VERSION, VERSION)
        return shm

    @classmethod
    def open(cls, path: str) -> "WorkerShm":
        fd = os.open(path, os.O_RDWR)
        mm = mmap.mmap(fd, SHM_SIZE)
        shm = cls(path, mm, fd)
        if shm.read_u32(OFF_MAGIC) != MAGIC:
            raise ValueError(f"{path}: bad shm magic")
        return shm

    def close(self):
        self._mm.close()
        os.close(self._fd)

    # ------------------------------------------------------- raw accessors

    def read_u32(self, off: int) -> int:
        return _u32.unpack_from(self._mm, off)[0]

    def write_u32(self, off: int, v: int) -> None:
        _u32.pack_into(self._mm, off, v & 0xFFFFFFFF)

    def read_u64(self, off: int) -> int:
        return _u64.unpack_from(self._mm, off)[0]

    def write_u64(self, off: int, v: int) -> None:
        _u64.pack_into(self._mm, off, v & 0xFFFFFFFFFFFFFFFF)

    def read_f64(self, off: int) -> float:
        return _f64.unpack_from(self._mm, off)[0]

    def write_f64(self, off: int, v: float) -> None:
        _f64.pack_into(self._mm, off, v)

    # ---------------------------------------------------------- device ops

    def _dev_off(self, i: int) -> int:
        if not 0 <= i < MAX_DEVICES:
            raise IndexError(i)
        return OFF_DEV + i * DEV_STRIDE

    def set_device(self, i: int, uuid: str, up_limit_percent: int,
                   mem_limit_bytes: int, total_cus: int = 256,
                   refill_rate: float = 1000.0, capacity: float = 2000.0):
        off = self._dev_off(i)
        raw = uuid.encode()[:UUID_LEN - 1]
        self._mm[off:off + UUID_LEN] = raw + b"\x00" * (UUID_LEN - len(raw))
        self.write_u32(off + E_UP_LIMIT, up_limit_percent)
        self.write_u32(off + E_TOTAL_CUS, total_cus)
        self.write_u64(off + E_MEM_LIMIT, mem_limit_bytes)
        self.write_f64(off + E_RATE, refill_rate)
        self.write_f64(off + E_CAPACITY, capacity)
        self.write_f64(off + E_TOKENS, capacity)
        self.write_u64(off + E_LAST_UPDATE, time.monotonic_ns())
        self.write_u32(off + E_ACTIVE, 1)
        n = self.read_u32(OFF_DEVICE_COUNT)
        if i >= n:
            self.write_u32(OFF_DEVICE_COUNT, i + 1)

    def update_erl(self, i: int, refill_rate: float, capacity: float):
        off = self._dev_off(i)
        self.write_f64(off + E_RATE, refill_rate)
        self.write_f64(off + E_CAPACITY, capacity)

    def set_limits(self, i: int, up_limit_percent: Optional[int] = None,
                   mem_limit_bytes: Optional[int] = None):
        off = self._dev_off(i)
        if up_limit_percent is not None:
            self.write_u32(off + E_UP_LIMIT, up_limit_percent)
        if mem_limit_bytes is not None:
            self.write_u64(off + E_MEM_LIMIT, mem_limit_bytes)

    def device(self, i: int) -> DeviceEntrySnapshot:
        off = self._dev_off(i)
        raw = bytes(self._mm[off:off + UUID_LEN])
        return DeviceEntrySnapshot(
            uuid=raw.split(b"\x00", 1)[0].decode(errors="replace"),
            up_limit_percent=self.read_u32(off + E_UP_LIMIT),
            total_cus=self.read_u32(off + E_TOTAL_CUS),
            mem_limit_bytes=self.read_u64(off + E_MEM_LIMIT),
            pod_memory_used=self.read_u64(off + E_MEM_USED),
            erl_refill_rate=self.read_f64(off + E_RATE),
            erl_capacity=self.read_f64(off + E_CAPACITY),
            erl_tokens=self.read_f64(off + E_TOKENS),
            erl_last_update_ns=self.read_u64(off + E_LAST_UPDATE),
            active=bool(self.read_u32(off + E_ACTIVE)),
            launch_count=self.read_u32(off + E_LAUNCHES),
            block_ns_total=self.read_u64(off + E_BLOCK_NS),
            alloc_bytes_total=self.read_u64(off + E_ALLOC_BYTES),
            vmm_bytes=self.read_u64(off + E_VMM_BYTES),
        )

    def devices(self) -> List[DeviceEntrySnapshot]:
        return [self.device(i) for i in range(self.read_u32(OFF_DEVICE_COUNT))]

    # ---------------------------------------------------------------- pids

    def _lock(self):
        # Cross-process spin mutex; Python holders are short (pid list edits).
        deadline = time.monotonic() + 2.0
        while True:
            if self.read_u32(OFF_MUTEX) == 0:
                # non-atomic CAS is fine between Python writers (hypervisor is
                # single-threaded for pid edits); the C++ side uses real CAS
                # and only ever spins, never writes pids without the lock.
                self.write_u32(OFF_MUTEX, 1)
                if self.read_u32(OFF_MUTEX) == 1:
                    return
            if time.monotonic() > deadline:  # stale holder crashed: steal
                self.write_u32(OFF_MUTEX, 1)
                return
            time.sleep(0.0005)

    def _unlock(self):
        self.write_u32(OFF_MUTEX, 0)

    def pids(self) -> List[int]:
        n = min(self.read_u32(OFF_PID_COUNT), MAX_PIDS)
        return [struct.unpack_from("<i", self._mm, OFF_PIDS + 4 * k)[0]
                for k in range(n)]

    def add_pid(self, pid: int):
        self._lock()
        try:
            cur = self.pids()
            if pid in cur:
                return
            if len(cur) >= MAX_PIDS:
                raise OverflowError("pid set full")
            struct.pack_into("<i", self._mm, OFF_PIDS + 4 * len(cur), pid)
            self.write_u32(OFF_PID_COUNT, len(cur) + 1)
        finally:
            self._unlock()

    def remove_pid(self, pid: int):
        self._lock()
        try:
            cur = [p for p in self.pids() if p != pid]
            for k, p in enumerate(cur):
                struct.pack_into("<i", self._mm, OFF_PIDS + 4 * k, p)
            self.write_u32(OFF_PID_COUNT, len(cur))
        finally:
            self._unlock()

    def sweep_dead_pids(self) -> List[int]:
        """Drop pids whose process is gone (reference isProcessAlive kill-0)."""

        dead = []
        for p in self.pids():
            try:
                os.kill(p, 0)
            except ProcessLookupError:
                dead.append(p)
            except PermissionError:
                pass
        for p in dead:
            self.remove_pid(p)
        return dead

    # ----------------------------------------------------- flags/heartbeat

    def heartbeat(self) -> int:
        return self.read_u64(OFF_HEARTBEAT)

    def touch_hypervisor_heartbeat(self):
        self.write_u64(OFF_HYP_HEARTBEAT, time.monotonic_ns())

    def flags(self) -> int:
        return self.read_u32(OFF_FLAGS)

    def set_flag(self, bit: int, on: bool):
        f = self.read_u32(OFF_FLAGS)
        self.write_u32(OFF_FLAGS, (f | bit) if on else (f & ~bit))

    def freeze(self, on: bool):
        self.set_flag(FLAG_FREEZE, on)
