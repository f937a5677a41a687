"""Remoting transport on CPU: ring correctness + URL parsing + bundle size
logic (GPU end-to-end lives in test_gpu_remoting.py)."""
import json
import os
import subprocess

import pytest

from tensor_fusion_amd.client.runtime import parse_connection_url


def test_ring_spsc(native_built):
    out = subprocess.run([os.path.join(native_built, "tf_ring_test"), "200000"],
                         capture_output=True, text=True, timeout=120)
    assert out.returncode == 0, out.stderr
    assert json.loads(out.stdout)["ok"] is True


def test_connection_url_roundtrip():
    from tensor_fusion_amd.api.types import format_connection_url
    url = format_connection_url("10.0.0.7", 8000, "pool-a-worker-3", 17)
    d = parse_connection_url(url)
    assert d["ip"] == "10.0.0.7"
    assert d["port"] == "8000"
    assert d["worker"] == "pool-a-worker-3"
    assert d["rev"] == "17"
    with pytest.raises(ValueError):
        parse_connection_url("http://nope")


def test_codeobj_signatures(native_built, tmp_path):
    """comgr parses kernarg layouts out of a gfx950 code object (the worker
    relies on this for hipLaunchKernel arg packing)."""

    hip_src = tmp_path / "k.hip"
    hip_src.write_text(
        '#include <hip/hip_runtime.h>\n'
        'extern "C" __global__ void axb(float a, const float* x, float* y,'
        ' int n) { int i = blockIdx.x * blockDim.x + threadIdx.x;'
        ' if (i < n) y[i] = a * x[i]; }\n')
    hsaco = tmp_path / "k.hsaco"
    subprocess.run(["/opt/rocm/bin/hipcc", "--genco",
                    "--offload-arch=gfx950", str(hip_src), "-o", str(hsaco)],
                   check=True, capture_output=True)
    # build the dump tool on demand
    dump = tmp_path / "codump"
    subprocess.run(["g++", "-O2", "-std=c++17", "-I", "native/remoting",
                    "native/remoting/codeobj_dump.cpp",
                    "native/remoting/codeobj.cpp", "-ldl", "-o", str(dump)],
                   check=True, cwd=os.path.dirname(os.path.dirname(
                       os.path.abspath(__file__))))
    out = subprocess.run([str(dump), str(hsaco)], capture_output=True,
                         text=True, check=True)
    sigs = json.loads(out.stdout)
    assert "axb" in sigs
    assert sigs["axb"]["explicit"] == 28
    assert sigs["axb"]["args"] == [[4, 0], [8, 8], [8, 16], [4, 24]]
