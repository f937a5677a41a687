"""GPU-over-IP remoting on a real MI355X.

The client process runs with HIP_VISIBLE_DEVICES="" — it cannot touch the
GPU except through the worker, so passing results prove genuine remoting.
"""
import json
import os
import subprocess
import sys

import pytest

pytestmark = pytest.mark.gpu

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
NATIVE = os.path.join(REPO, "tensor_fusion_amd", "_native")

sys.path.insert(0, REPO)
from tensor_fusion_amd.client.runtime import client_env, start_worker  # noqa


@pytest.fixture()
def worker(tmp_path):
    h = start_worker(str(tmp_path / "vgpu.sock"), device_index=0)
    yield h
    h.stop()


def test_testapp_native_baseline():
    out = subprocess.run([os.path.join(NATIVE, "tf_remote_testapp")],
                         capture_output=True, text=True, timeout=120)
    assert "TESTAPP_OK" in out.stdout, out.stdout + out.stderr


def test_testapp_remoted(worker):
    env = client_env(worker.socket_path)
    out = subprocess.run([os.path.join(NATIVE, "tf_remote_testapp")],
                         capture_output=True, text=True, timeout=180, env=env)
    assert "TESTAPP_OK" in out.stdout, out.stdout + out.stderr
    assert "devices=1" in out.stdout
    # the vGPU is a full MI355X: 288 GB visible through the remote path
    assert "vram_total_gb=" in out.stdout


def test_torch_tiny_decode_remoted(worker, tmp_path):
    """Full PyTorch through the remoting path: tiny Llama decode."""

    env = client_env(worker.socket_path)
    env["PYTORCH_NO_HIP_MEMORY_CACHING"] = env.get(
        "PYTORCH_NO_HIP_MEMORY_CACHING", "0")
    out = subprocess.run(
        [sys.executable, "-m", "tensor_fusion_amd.models.llama", "--model",
         "tiny", "--batch", "2", "--ctx", "16", "--steps", "4", "--warmup",
         "1"],
        capture_output=True, text=True, timeout=600, env=env, cwd=REPO)
    assert out.returncode == 0, out.stdout[-2000:] + out.stderr[-4000:]
    r = json.loads(out.stdout.strip().splitlines()[-1])
    assert r["tok_s"] > 0
