#!/usr/bin/env python3
"""Builds every native artifact in-tree (tensor_fusion_amd/_native/).

Called by __graft_entry__.build() and usable standalone:
    python build_native.py [--only NAME]

Artifacts:
  libtfhip_limiter.so   LD_PRELOAD soft-isolation limiter (g++, no HIP dep)
  libmockhip.so         CPU mock of the HIP entry points (CI)
  tf_limiter_testbed    limiter test driver (links libmockhip)
  tf_shm_layout_dump    shm layout cross-check helper
  libaccelerator_amd.so device discovery/metrics ABI over amd-smi (g++)
  tf_accel_dump         accelerator ABI test driver (dlopen)
  libtfremote.so + tf_vgpu_worker  GPU-over-IP remoting (g++, dlopens HIP)
  libtftier.so          VRAM-tiering HIP kernels (hipcc --offload-arch=gfx950)

The .so files are git-ignored but DO travel to the GPU box with gpurun.
"""
from __future__ import annotations

import os
import subprocess
import sys

ROOT = os.path.dirname(os.path.abspath(__file__))
NATIVE = os.path.join(ROOT, "native")
OUT = os.path.join(ROOT, "tensor_fusion_amd", "_native")

CXX = os.environ.get("CXX", "g++")
HIPCC = os.environ.get("HIPCC", "/opt/rocm/bin/hipcc")
ROCM = os.environ.get("ROCM_PATH", "/opt/rocm")

CXXFLAGS = ["-O2", "-g", "-std=c++17", "-fPIC", "-Wall"]


def _newer(target: str, sources: list) -> bool:
    if not os.path.exists(target):
        return True
    t = os.path.getmtime(target)
    return any(os.path.getmtime(s) > t for s in sources if os.path.exists(s))


def _run(cmd: list):
    print("+ " + " ".join(cmd), flush=True)
    subprocess.run(cmd, check=True, cwd=ROOT)


def _cc(out: str, sources: list, extra: list = None, shared: bool = True):
    os.makedirs(OUT, exist_ok=True)
    target = os.path.join(OUT, out)
    deps = sources + [os.path.join(NATIVE, "limiter", "limiter_shm.h")]
    if not _newer(target, deps):
        return target
    cmd = [CXX] + CXXFLAGS + ["-I", os.path.join(NATIVE, "limiter")]
    if shared:
        cmd += ["-shared"]
    cmd += sources + ["-o", target] + (extra or [])
    _run(cmd)
    return target


def build_limiter():
    return _cc("libtfhip_limiter.so",
               [os.path.join(NATIVE, "limiter", "hip_limiter.cpp")],
               extra=["-ldl", "-pthread"])


def build_mockhip():
    return _cc("libmockhip.so", [os.path.join(NATIVE, "mock", "mock_hip.cpp")],
               extra=["-pthread"])


def build_testbed():
    build_mockhip()
    return _cc("tf_limiter_testbed",
               [os.path.join(NATIVE, "mock", "limiter_testbed.cpp")],
               extra=["-L", OUT, "-lmockhip", "-ldl",
                      f"-Wl,-rpath,$ORIGIN"],
               shared=False)


def build_shm_dump():
    return _cc("tf_shm_layout_dump",
               [os.path.join(NATIVE, "limiter", "shm_layout_dump.cpp")],
               shared=False)


def build_accelerator():
    src = os.path.join(NATIVE, "accelerator", "accelerator_amd.cpp")
    if not os.path.exists(src):
        return None
    return _cc("libaccelerator_amd.so", [src],
               extra=["-I", os.path.join(ROCM, "include"),
                      "-I", os.path.join(NATIVE, "accelerator"),
                      "-ldl", "-pthread"])


def build_accel_dump():
    src = os.path.join(NATIVE, "accelerator", "accel_dump.cpp")
    if not os.path.exists(src):
        return None
    return _cc("tf_accel_dump", [src],
               extra=["-I", os.path.join(NATIVE, "accelerator"), "-ldl"],
               shared=False)


def build_remoting():
    cdir = os.path.join(NATIVE, "remoting")
    client = os.path.join(cdir, "client_interpose.cpp")
    worker = os.path.join(cdir, "worker_main.cpp")
    out = []
    if os.path.exists(client):
        out.append(_cc("libtfhip_client.so", [client],
                       extra=["-I", cdir, "-ldl", "-pthread"]))
    if os.path.exists(worker):
        out.append(_cc("tf_vgpu_worker",
                       [worker, os.path.join(cdir, "codeobj.cpp")],
                       extra=["-I", cdir, "-ldl", "-pthread"], shared=False))
    probe = os.path.join(cdir, "vmm_probe.cpp")
    if os.path.exists(probe):
        out.append(_cc("tf_vmm_probe", [probe],
                       extra=["-I", cdir, "-ldl"], shared=False))
    ring_test = os.path.join(cdir, "ring_test.cpp")
    if os.path.exists(ring_test):
        out.append(_cc("tf_ring_test", [ring_test],
                       extra=["-I", cdir, "-pthread"], shared=False))
    testapp = os.path.join(cdir, "test_app.hip")
    if os.path.exists(testapp):
        target = os.path.join(OUT, "tf_remote_testapp")
        if _newer(target, [testapp]):
            _run([HIPCC, "--offload-arch=gfx950", "-O2", "-std=c++17",
                  testapp, "-o", target])
        out.append(target)
    return out


def build_tiering():
    """HIP kernels: cross-compiled for gfx950 (no GPU needed to build)."""

    src = os.path.join(NATIVE, "tiering", "tier_kernels.hip")
    if not os.path.exists(src):
        return None
    os.makedirs(OUT, exist_ok=True)
    target = os.path.join(OUT, "libtftier.so")
    if not _newer(target, [src]):
        return target
    _run([HIPCC, "--offload-arch=gfx950", "-O3", "-std=c++17", "-fPIC",
          "-shared", src, "-o", target])
    return target


def build_bw_sweep():
    src = os.path.join(NATIVE, "tiering", "bw_sweep.hip")
    if not os.path.exists(src):
        return None
    target = os.path.join(OUT, "tf_bw_sweep")
    if not _newer(target, [src]):
        return target
    _run([HIPCC, "--offload-arch=gfx950", "-O3", "-std=c++17", src, "-o",
          target])
    return target


def build_fused_ops():
    """Fused normalization kernels (gfx950)."""

    srcs = [os.path.join(NATIVE, "ops", "fused_ops.hip"),
            os.path.join(NATIVE, "ops", "skinny_gemm.hip")]
    srcs = [s0 for s0 in srcs if os.path.exists(s0)]
    if not srcs:
        return None
    os.makedirs(OUT, exist_ok=True)
    target = os.path.join(OUT, "libtfops.so")
    if not _newer(target, srcs):
        return target
    _run([HIPCC, "--offload-arch=gfx950", "-O3", "-std=c++17", "-fPIC",
          "-shared"] + srcs + ["-o", target])
    return target


ALL = {
    "limiter": build_limiter,
    "mockhip": build_mockhip,
    "testbed": build_testbed,
    "shm_dump": build_shm_dump,
    "accelerator": build_accelerator,
    "accel_dump": build_accel_dump,
    "remoting": build_remoting,
    "tiering": build_tiering,
    "fused_ops": build_fused_ops,
    "bw_sweep": build_bw_sweep,
}


def build_all(only=None):
    results = {}
    for name, fn in ALL.items():
        if only and name != only:
            continue
        results[name] = fn()
    return results


if __name__ == "__main__":
    only = None
    if "--only" in sys.argv:
        only = sys.argv[sys.argv.index("--only") + 1]
    build_all(only)
    print("native build OK")
