"""Shm protocol: C++/Python layout agreement + Python-side operations."""
import json
import os
import subprocess

from tensor_fusion_amd.hypervisor import shm as S


def test_layout_matches_cpp(native_built, tmp_path):
    out = subprocess.run([os.path.join(native_built, "tf_shm_layout_dump")],
                         capture_output=True, text=True, check=True)
    d = json.loads(out.stdout)
    assert d["size"] == S.SHM_SIZE
    assert d["magic"] == S.MAGIC
    assert d["version"] == S.VERSION
    assert d["dev_off"] == S.OFF_DEV
    assert d["dev_stride"] == S.DEV_STRIDE
    assert d["device_count_off"] == S.OFF_DEVICE_COUNT
    assert d["flags_off"] == S.OFF_FLAGS
    assert d["heartbeat_off"] == S.OFF_HEARTBEAT
    assert d["hyp_heartbeat_off"] == S.OFF_HYP_HEARTBEAT
    assert d["mutex_off"] == S.OFF_MUTEX
    assert d["pid_count_off"] == S.OFF_PID_COUNT
    assert d["pids_off"] == S.OFF_PIDS
    assert d["e_uuid"] == S.E_UUID
    assert d["e_up_limit"] == S.E_UP_LIMIT
    assert d["e_total_cus"] == S.E_TOTAL_CUS
    assert d["e_mem_limit"] == S.E_MEM_LIMIT
    assert d["e_mem_used"] == S.E_MEM_USED
    assert d["e_rate"] == S.E_RATE
    assert d["e_capacity"] == S.E_CAPACITY
    assert d["e_tokens"] == S.E_TOKENS
    assert d["e_last_update"] == S.E_LAST_UPDATE
    assert d["e_active"] == S.E_ACTIVE
    assert d["e_launches"] == S.E_LAUNCHES
    assert d["e_block_ns"] == S.E_BLOCK_NS
    assert d["e_alloc_bytes"] == S.E_ALLOC_BYTES


def test_create_and_roundtrip(tmp_path):
    p = str(tmp_path / "ns" / "pod" / "shm")
    w = S.WorkerShm.create(p)
    w.set_device(0, "GPU-abc-123", up_limit_percent=25,
                 mem_limit_bytes=8 << 30, refill_rate=1234.5, capacity=100.0)
    d = w.device(0)
    assert d.uuid == "GPU-abc-123"
    assert d.up_limit_percent == 25
    assert d.mem_limit_bytes == 8 << 30
    assert abs(d.erl_refill_rate - 1234.5) < 1e-9
    assert d.active
    assert len(w.devices()) == 1

    # open-not-truncate: re-create preserves state
    w.write_u64(S.OFF_DEV + S.E_MEM_USED, 4096)
    w.close()
    w2 = S.WorkerShm.create(p)
    assert w2.device(0).pod_memory_used == 4096
    w2.close()


def test_pid_set(tmp_path):
    w = S.WorkerShm.create(str(tmp_path / "shm"))
    w.add_pid(os.getpid())
    w.add_pid(os.getpid())  # idempotent
    w.add_pid(999999999)  # definitely dead
    assert set(w.pids()) == {os.getpid(), 999999999}
    dead = w.sweep_dead_pids()
    assert dead == [999999999]
    assert w.pids() == [os.getpid()]
    w.remove_pid(os.getpid())
    assert w.pids() == []
    w.close()


def test_flags_and_erl_update(tmp_path):
    w = S.WorkerShm.create(str(tmp_path / "shm"))
    w.set_device(0, "u", 50, 1 << 30)
    w.freeze(True)
    assert w.flags() & S.FLAG_FREEZE
    w.freeze(False)
    assert not (w.flags() & S.FLAG_FREEZE)
    w.update_erl(0, 777.0, 42.0)
    d = w.device(0)
    assert d.erl_refill_rate == 777.0
    assert d.erl_capacity == 42.0
    w.close()
