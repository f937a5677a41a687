#!/usr/bin/env python3
"""Fake `amd-smi` — vGPU-scoped device visibility inside workload pods.

Reference: a fake nvidia-smi is bind-mounted over /usr/local/bin/nvidia-smi
in workload containers (utils/compose.go:141-145, :464-470) so `nvidia-smi`
shows the pod its vGPU limits instead of the whole device. This is the
MI355X equivalent: installed as `amd-smi` (and `rocm-smi`) in the pod, it
reads the limiter shm (TF_SHM_PATH) — or the TF_* env in standalone mode —
and prints the vGPU's capacity/usage in amd-smi-like form.

Usage: fake_amd_smi.py [list|metric|monitor] (all print the same summary).
"""
from __future__ import annotations

import json
import os
import sys


def read_shm(path):
    """Parse the limiter shm page (layout: native/limiter/limiter_shm.h)."""

    import struct
    with open(path, "rb") as f:
        raw = f.read(4096)
    magic, version = struct.unpack_from("<II", raw, 0)
    if magic != 0x5446414D:
        return None
    devs = []
    for i in range(16):
        off = 8 + i * 160
        uuid = raw[off:off + 64].split(b"\0")[0].decode(errors="replace")
        (up_limit, total_cus, mem_limit, mem_used, rate_bits, cap_bits,
         tok_bits, _last, active, launches, block_ns, alloc_total
         ) = struct.unpack_from("<IIQQQQQQIIQQ", raw, off + 64)
        if not active:
            continue
        import struct as _s
        devs.append({
            "uuid": uuid or f"vgpu-{i}",
            "up_limit_percent": up_limit,
            "total_cus": total_cus,
            "mem_limit": mem_limit,
            "mem_used": mem_used,
            "erl_rate": _s.unpack("<d", _s.pack("<Q", rate_bits))[0],
            "erl_tokens": _s.unpack("<d", _s.pack("<Q", tok_bits))[0],
            "launches": launches,
            "block_ns": block_ns,
        })
    return devs


def devices_from_env():
    lim = int(os.environ.get("TF_VRAM_LIMIT_BYTES", 0))
    up = int(float(os.environ.get("TF_UP_LIMIT_PERCENT", 100)))
    return [{
        "uuid": "standalone-0", "up_limit_percent": up, "total_cus": 256,
        "mem_limit": lim, "mem_used": 0, "erl_rate": 0.0, "erl_tokens": 0.0,
        "launches": 0, "block_ns": 0,
    }]


def main(argv=None):
    argv = argv if argv is not None else sys.argv[1:]
    shm_path = os.environ.get("TF_SHM_PATH", "")
    devs = None
    if shm_path and os.path.exists(shm_path):
        try:
            devs = read_shm(shm_path)
        except Exception:
            devs = None
    if devs is None:
        devs = devices_from_env()

    if "--json" in argv:
        print(json.dumps({"gpus": devs}))
        return 0

    print("TENSOR-FUSION vGPU (MI355X)            amd-smi compatible view")
    print("=" * 64)
    for i, d in enumerate(devs):
        vram_total_gb = d["mem_limit"] / (1 << 30) if d["mem_limit"] \
            else 288.0
        vram_used_gb = d["mem_used"] / (1 << 30)
        cus = int(d["total_cus"] * d["up_limit_percent"] / 100)
        print(f"GPU[{i}]  {d['uuid']}")
        print(f"  VRAM       : {vram_used_gb:8.2f} / {vram_total_gb:.2f} GiB")
        print(f"  COMPUTE    : {d['up_limit_percent']}% of device "
              f"({cus}/{d['total_cus']} CUs)")
        print(f"  ERL        : rate={d['erl_rate']:.0f} tok/s "
              f"bucket={d['erl_tokens']:.1f}")
        print(f"  LAUNCHES   : {d['launches']}  "
              f"throttled={d['block_ns'] / 1e6:.1f} ms")
    return 0


if __name__ == "__main__":
    sys.exit(main())
