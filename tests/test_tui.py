"""TUI frames + key-navigation state machine (reference
pkg/hypervisor/tui: device/worker/metrics views, shm inspector dialog)."""
import time

from rich.console import Console

from tensor_fusion_amd.tui.app import (UiState, VIEWS, build_frame,
                                       shm_dialog_frame, summarize_metrics)

DEVICES = [{"index": 0, "uuid": "GPU-abc", "vram_total": 288 << 30,
            "vram_used": 100 << 30, "busy_percent": 42.0,
            "compute_units": 256, "compute_partition": "SPX",
            "worker_count": 2}]
WORKERS = [
    {"namespace": "default", "pod": f"w{i}", "qos": "medium",
     "isolation": "soft", "shm_path": f"/run/tf/default/w{i}/shm",
     "device_uuid": "GPU-abc", "heartbeat_ts": time.time(),
     "limits": {"vram": 96 << 30, "compute_percent": 25},
     "usage": {"vram": (20 + i) << 30, "erl_rate": 1500.0,
               "erl_tokens": 42.0, "erl_capacity": 150.0,
               "launches": 1000 * (i + 1), "block_ns": 5e8,
               "vmm_bytes": 1 << 30},
     "pids": [100 + i], "flags": 2 if i == 1 else 0}
    for i in range(3)]


def render(state: UiState) -> str:
    c = Console(record=True, width=130)
    c.print(build_frame(state))
    return c.export_text()


def mk_state(view="devices"):
    s = UiState(view=view, devices=DEVICES, workers=WORKERS,
                metrics=summarize_metrics(DEVICES, WORKERS),
                tier=[{"pod": "default/w1", "budget": 70 << 30,
                       "resident": 60 << 30, "host_bytes": 17 << 30,
                       "demoted": 4 << 30, "promoted": 2 << 30,
                       "pressured": True}])
    return s


class TestFrames:
    def test_devices_view(self):
        out = render(mk_state("devices"))
        assert "GPU-abc" in out and "SPX" in out and "288" in out

    def test_workers_view(self):
        out = render(mk_state("workers"))
        assert "default/w0" in out and "1500" in out

    def test_metrics_view(self):
        out = render(mk_state("metrics"))
        assert "launches" in out and "6000" in out  # 1000+2000+3000

    def test_tier_view(self):
        out = render(mk_state("tier"))
        assert "default/w1" in out and "YES" in out  # pressured

    def test_shm_inspector(self):
        c = Console(record=True, width=130)
        c.print(shm_dialog_frame(WORKERS[0]))
        out = c.export_text()
        assert "shm inspector" in out
        assert "/run/tf/default/w0/shm" in out
        assert "erl_refill_rate" in out
        assert "vmm_bytes" in out


class TestKeyMachine:
    def test_view_switching(self):
        s = mk_state()
        for key, view in (("2", "workers"), ("4", "tier"),
                          ("1", "devices")):
            assert s.handle_key(key)
            assert s.view == view
        s.handle_key("\t")
        assert s.view == "workers"

    def test_selection_clamped(self):
        s = mk_state("workers")
        for _ in range(10):
            s.handle_key("j")
        assert s.selected == len(WORKERS) - 1
        for _ in range(10):
            s.handle_key("k")
        assert s.selected == 0

    def test_enter_opens_and_q_closes_dialog(self):
        s = mk_state("workers")
        s.handle_key("j")
        s.handle_key("\n")
        assert s.shm_dialog is not None
        assert s.shm_dialog["pod"] == "w1"
        out = render(s)
        assert "shm inspector" in out
        assert s.handle_key("q")  # closes dialog, does NOT quit
        assert s.shm_dialog is None
        assert s.handle_key("q") is False  # now quits

    def test_help_overlay(self):
        s = mk_state()
        s.handle_key("?")
        assert s.show_help
        assert "switch view" in render(s)
        s.handle_key("x")  # any key closes
        assert not s.show_help
