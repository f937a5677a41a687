"""LD_PRELOAD limiter over the mock HIP runtime (CPU-only CI).

GPU behaviour (real hipMalloc/hipLaunchKernel) is covered by
tests/test_gpu_limiter.py (@gpu).
"""
import json
import os
import subprocess

import pytest

from tensor_fusion_amd.hypervisor import shm as S


def run_testbed(native, scenario, *args, env=None, preload=True):
    e = dict(os.environ)
    e.pop("TF_SHM_PATH", None)
    if preload:
        e["LD_PRELOAD"] = os.path.join(native, "libtfhip_limiter.so")
    e.update(env or {})
    out = subprocess.run(
        [os.path.join(native, "tf_limiter_testbed"), scenario, *map(str, args)],
        capture_output=True, text=True, env=e, timeout=60)
    assert out.returncode == 0, out.stderr
    return json.loads(out.stdout.strip().splitlines()[-1])


def test_passthrough_without_config(native_built):
    r = run_testbed(native_built, "alloc", 4, 1 << 20)
    assert r == {"ok": 4, "denied": 0, "readmitted": 0}


def test_vram_cap(native_built):
    r = run_testbed(native_built, "alloc", 12, 1 << 20,
                    env={"TF_VRAM_LIMIT_BYTES": str(8 << 20)})
    assert r["ok"] == 8
    assert r["denied"] == 4
    assert r["readmitted"] == 1  # free frees budget again


def test_erl_throttle_rate(native_built):
    r = run_testbed(native_built, "launch", 300,
                    env={"TF_UP_LIMIT_PERCENT": "25", "TF_ERL_RATE": "1000",
                         "TF_ERL_CAPACITY": "10"})
    # 300 launches at 1000 tokens/s ≈ 0.3 s (capacity gives a small head start)
    assert 0.2 < r["elapsed_s"] < 1.0, r


def test_unthrottled_at_100_percent(native_built):
    r = run_testbed(native_built, "launch", 2000,
                    env={"TF_UP_LIMIT_PERCENT": "100"})
    assert r["elapsed_s"] < 0.5  # bucket bypassed entirely


def test_hypervisor_shm_governs_limiter(native_built, tmp_path):
    """End-to-end: hypervisor-side Python writes the shm page, the C++
    limiter in a separate process obeys it and reports stats back."""

    p = str(tmp_path / "shm")
    w = S.WorkerShm.create(p)
    w.set_device(0, "GPU-e2e", up_limit_percent=25, mem_limit_bytes=4 << 20,
                 refill_rate=500.0, capacity=5.0)

    env = {"TF_SHM_PATH": p}
    r = run_testbed(native_built, "alloc", 8, 1 << 20, env=env)
    assert r["ok"] == 4 and r["denied"] == 4

    r = run_testbed(native_built, "launch", 200, env=env)
    assert 0.3 < r["elapsed_s"] < 1.2  # 200/500 = 0.4s expected

    # limiter wrote usage stats back into the page
    d = w.device(0)
    assert d.launch_count >= 200
    assert d.block_ns_total > 0
    assert w.heartbeat() > 0
    # pid was registered and is now dead => sweep removes it
    assert len(w.pids()) >= 1
    w.sweep_dead_pids()
    assert w.pids() == []
    w.close()


def test_freeze_blocks_and_resumes(native_built, tmp_path):
    p = str(tmp_path / "shm")
    w = S.WorkerShm.create(p)
    w.set_device(0, "GPU-frz", up_limit_percent=50, mem_limit_bytes=0,
                 refill_rate=1e9, capacity=1e9)
    w.freeze(True)

    env = dict(os.environ)
    env["LD_PRELOAD"] = os.path.join(native_built, "libtfhip_limiter.so")
    env["TF_SHM_PATH"] = p
    proc = subprocess.Popen(
        [os.path.join(native_built, "tf_limiter_testbed"), "launch", "10"],
        env=env, stdout=subprocess.PIPE, text=True)
    try:
        proc.wait(timeout=0.5)
        pytest.fail("frozen worker should not finish")
    except subprocess.TimeoutExpired:
        pass
    w.freeze(False)
    out, _ = proc.communicate(timeout=10)
    r = json.loads(out.strip())
    assert r["launches"] == 10
    w.close()


def test_latency_histogram_traces_launches(native_built):
    r = run_testbed(native_built, "lat", 250,
                    env={"TF_LIMITER_TRACE": "1",
                         "TF_UP_LIMIT_PERCENT": "100"})
    assert r["launch_samples"] == 250


def test_fake_amd_smi_reads_shm(tmp_path):
    """tools/fake_amd_smi shows the vGPU-scoped view from the limiter shm
    (reference: fake nvidia-smi bind-mounted into workload pods)."""

    import json as _json
    import subprocess as _sp
    import sys as _sys

    from tensor_fusion_amd.hypervisor import shm as S
    path = str(tmp_path / "shm")
    page = S.WorkerShm.create(path)
    page.set_device(0, "vgpu-abc", up_limit_percent=25,
                    mem_limit_bytes=8 << 30)
    env = dict(os.environ)
    env["TF_SHM_PATH"] = path
    out = _sp.run([_sys.executable, "-m",
                   "tensor_fusion_amd.tools.fake_amd_smi", "--json"],
                  env=env, capture_output=True, text=True, timeout=60)
    assert out.returncode == 0, out.stderr
    data = _json.loads(out.stdout)
    assert data["gpus"][0]["uuid"] == "vgpu-abc"
    assert data["gpus"][0]["up_limit_percent"] == 25
    assert data["gpus"][0]["mem_limit"] == 8 << 30
