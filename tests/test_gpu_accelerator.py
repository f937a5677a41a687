"""libaccelerator_amd.so against the REAL amd-smi on an MI355X box
(everything else runs on the mock; this validates the production path —
reference provider/test/test_accelerator.c is the analog)."""
import os
import sys

import pytest

pytestmark = pytest.mark.gpu

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)


@pytest.fixture(scope="module")
def accel():
    env_backup = os.environ.pop("TF_ACCEL_MOCK", None)
    from tensor_fusion_amd.hypervisor.device import Accelerator
    a = Accelerator()
    yield a
    if env_backup is not None:
        os.environ["TF_ACCEL_MOCK"] = env_backup


def test_real_device_discovery(accel):
    devs = accel.devices()
    assert len(devs) >= 1
    d = devs[0]
    # MI355X facts: 288 GB HBM3E, 256 CUs
    assert d.vram_total > 280 * 1024**3, d
    assert d.compute_units == 256, d
    assert d.uuid


def test_real_metrics(accel):
    m = accel.metrics(0)
    assert m.vram_total > 280 * 1024**3
    assert 0 <= m.gfx_activity <= 100


def test_topology_self_tier(accel):
    n = len(accel.devices())
    topo = accel.topology(n)
    for i in range(n):
        assert topo[i][i] == 0  # self = tier 0


def test_hypervisor_stack_on_real_device(accel, tmp_path):
    """Device controller + worker shm + ERL against the real device."""

    from tensor_fusion_amd.hypervisor.main import build_hypervisor
    devices, workers, erl, _ = build_hypervisor(
        shm_root=str(tmp_path / "shm"))
    devs = devices.devices()
    assert devs and devs[0].vram_total > 280 * 1024**3


def test_partition_cu_mask_on_hardware(accel):
    """XCD-slab partition → HSA_CU_MASK composed by the accelerator lib
    actually confines a workload: 1 XCD (32 CUs) runs compute-bound work
    measurably slower than the full device (hard partition path,
    reference AssignPartition → MI355X CPX-slab design)."""

    import json
    import subprocess
    import sys

    env_str = accel.cu_mask_env_for_xcds(0, [0])  # XCD 0 only
    assert env_str.startswith("HSA_CU_MASK=")
    mask_val = env_str.split("=", 1)[1]

    child = r"""
import json, sys, time
import torch
a = torch.randn(4096, 4096, device="cuda", dtype=torch.bfloat16)
b = torch.randn(4096, 4096, device="cuda", dtype=torch.bfloat16)
for _ in range(10):
    a @ b
torch.cuda.synchronize()
t0 = time.perf_counter()
for _ in range(100):
    a @ b
torch.cuda.synchronize()
print(json.dumps({"s": time.perf_counter() - t0}))
"""

    def run(extra):
        import os as _os
        env = dict(_os.environ)
        env.update(extra)
        out = subprocess.run([sys.executable, "-c", child], env=env,
                             capture_output=True, text=True, timeout=300)
        assert out.returncode == 0, out.stderr[-1500:]
        return json.loads(out.stdout.strip().splitlines()[-1])["s"]

    # best-of-2 per config: absorbs one-off clock ramps / autotune noise
    full = min(run({}) for _ in range(2))
    part = min(run({"HSA_CU_MASK": mask_val}) for _ in range(2))
    assert part > 1.3 * full, (full, part)


def test_compute_partition_mode_api(accel):
    """AMD compute/memory partition-mode APIs against real amd-smi.
    Reads the device-global mode; exercises SET by writing back the
    CURRENT mode (a no-op change — flipping a shared pool box into CPX
    would renumber its devices for later tenants). Skips when the
    amd-smi build predates the partition APIs."""

    mode = accel.compute_partition(0)
    if mode is None:
        import pytest
        pytest.skip("amd-smi build lacks compute-partition APIs")
    assert mode in ("SPX", "DPX", "TPX", "QPX", "CPX"), mode
    mem = accel.memory_partition(0)
    if mem is not None:
        assert mem.startswith("NPS"), mem
    # no-op set: same mode back — validates the set path end to end
    ok = accel.set_compute_partition(0, mode)
    assert ok in (True, False)
    assert accel.compute_partition(0) == mode
