"""Auto-freeze TTL controller (reference AutoFreezeConfig
http_types.go:85-91; the reference's execution lives in closed vgpu.rs —
here the hypervisor freezes idle workers to disk itself and auto-resumes
on the client's next dial via a parked socket listener)."""
import os
import socket
import time

from tensor_fusion_amd.hypervisor.vgpu_manager import (AutoFreezeController,
                                                       VgpuWorker,
                                                       VgpuWorkerManager)


class _FakeProc:
    def poll(self):
        return None


class _FakeHandle:
    proc = _FakeProc()


class FakeManager(VgpuWorkerManager):
    def __init__(self, tmp):
        self.run_dir = str(tmp)
        self.workers = {}
        self.snapshots = []
        self.resumes = []

    def add(self, key):
        sock = os.path.join(self.run_dir, key.replace("/", "_") + ".sock")
        w = VgpuWorker(key=key, socket_path=sock, device_index=0,
                       snapshot_path=sock + ".snap")
        w.handle = _FakeHandle()
        self.workers[key] = w
        return w

    def snapshot(self, key):
        self.snapshots.append(key)
        return self.workers[key].snapshot_path

    def resume(self, key, device_index=None):
        self.resumes.append(key)
        return self.workers[key]


def test_freeze_after_idle_ttl_and_resume_on_dial(tmp_path):
    mgr = FakeManager(tmp_path)
    w = mgr.add("ns/w1")
    activity = {"v": 100.0}
    ctl = AutoFreezeController(
        mgr, rules={"low": {"enable": True, "freeze_to_disk_ttl_s": 5}},
        activity_fn=lambda key: activity["v"], activity_threshold=0.5)
    ctl.register("ns/w1", "low")

    t0 = time.time()
    ctl.tick(now=t0)             # baseline sample
    activity["v"] = 101.0        # busy
    ctl.tick(now=t0 + 2)
    ctl.tick(now=t0 + 4)         # idle 2s < ttl
    assert mgr.snapshots == []
    ctl.tick(now=t0 + 10)        # idle 6s >= ttl → freeze
    assert mgr.snapshots == ["ns/w1"]
    st = ctl.states["ns/w1"]
    assert st.phase == "frozen_disk" and st.freezes == 1
    assert os.path.exists(w.socket_path)  # parked listener

    # the client dials → auto-resume
    c = socket.socket(socket.AF_UNIX, socket.SOCK_STREAM)
    c.connect(w.socket_path)
    c.close()
    ctl.tick(now=t0 + 11)
    assert mgr.resumes == ["ns/w1"]
    assert st.phase == "active" and st.resumes == 1


def test_activity_resets_idle_clock(tmp_path):
    mgr = FakeManager(tmp_path)
    mgr.add("ns/w2")
    activity = {"v": 1.0}
    ctl = AutoFreezeController(
        mgr, rules={"low": {"enable": True, "freeze_to_disk_ttl_s": 5}},
        activity_fn=lambda key: activity["v"], activity_threshold=0.5)
    ctl.register("ns/w2", "low")
    t0 = time.time()
    for i in range(10):           # continuously busy for 20s
        activity["v"] += 1
        ctl.tick(now=t0 + 2 * i)
    assert mgr.snapshots == []


def test_disabled_rule_and_high_qos_untouched(tmp_path):
    mgr = FakeManager(tmp_path)
    mgr.add("ns/w3")
    ctl = AutoFreezeController(
        mgr, rules={"low": {"enable": True, "freeze_to_disk_ttl_s": 1},
                    "critical": {"enable": False}},
        activity_fn=lambda key: 7.0)
    ctl.register("ns/w3", "critical")
    t0 = time.time()
    ctl.tick(now=t0)
    ctl.tick(now=t0 + 100)
    assert mgr.snapshots == []
