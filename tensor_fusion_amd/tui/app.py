"""Hypervisor terminal UI — node operator's live dashboard.

Reference: pkg/hypervisor/tui/ + cmd/hypervisor-tui (bubbletea app over
the hypervisor HTTP API: device view, worker view, metrics, shm
inspector dialog). This is the rich-based equivalent: polls
http://<node>:8001/api/v1/{devices,workers} and renders device
utilization, per-worker vGPU limits/usage and ERL state; `--once` mode
prints a single frame (used by tests and headless boxes).
"""
from __future__ import annotations

import argparse
import json
import time
import urllib.request
from typing import Optional

from rich.console import Console, Group
from rich.panel import Panel
from rich.table import Table


def _fetch(base: str, path: str):
    with urllib.request.urlopen(f"{base}{path}", timeout=3) as r:
        return json.loads(r.read())


def build_frame(devices: list, workers: list):
    dt = Table(title="Devices (MI355X)", expand=True)
    for col in ("idx", "uuid", "VRAM used/total", "busy %", "CUs",
                "workers"):
        dt.add_column(col)
    for d in devices:
        used = d.get("vram_used", 0) / (1 << 30)
        total = d.get("vram_total", 0) / (1 << 30)
        dt.add_row(str(d.get("index", "?")), d.get("uuid", "")[:24],
                   f"{used:.1f}/{total:.0f} GiB",
                   f"{d.get('busy_percent', 0):.0f}",
                   str(d.get("compute_units", 256)),
                   str(d.get("worker_count", 0)))

    wt = Table(title="Workers (vGPUs)", expand=True)
    for col in ("pod", "qos", "isolation", "VRAM used/limit", "compute %",
                "ERL rate", "throttled ms", "hb age s"):
        wt.add_column(col)
    now = time.time()
    for w in workers:
        lim = w.get("limits", {})
        usage = w.get("usage", {})
        hb = w.get("heartbeat_ts", 0)
        wt.add_row(
            f"{w.get('namespace', '')}/{w.get('pod', '')}"[:32],
            w.get("qos", ""), w.get("isolation", ""),
            f"{usage.get('vram', 0) / (1 << 30):.1f}/"
            f"{lim.get('vram', 0) / (1 << 30):.1f} GiB",
            f"{lim.get('compute_percent', 100):.0f}",
            f"{usage.get('erl_rate', 0):.0f}",
            f"{usage.get('block_ns', 0) / 1e6:.0f}",
            f"{max(0, now - hb):.0f}" if hb else "-")
    return Group(Panel(dt), Panel(wt))


def render_once(base: str, console: Optional[Console] = None) -> str:
    console = console or Console(record=True, width=120)
    try:
        devices = _fetch(base, "/api/v1/devices").get("data", [])
    except Exception:
        devices = []
    try:
        workers = _fetch(base, "/api/v1/workers").get("data", [])
    except Exception:
        workers = []
    console.print(build_frame(devices, workers))
    return console.export_text() if console.record else ""


class HypervisorTUI:
    def __init__(self, base: str = "http://127.0.0.1:8001",
                 refresh_s: float = 2.0):
        self.base = base
        self.refresh_s = refresh_s

    def run(self):
        from rich.live import Live
        console = Console()
        with Live(console=console, refresh_per_second=2) as live:
            while True:
                try:
                    devices = _fetch(self.base, "/api/v1/devices").get(
                        "data", [])
                    workers = _fetch(self.base, "/api/v1/workers").get(
                        "data", [])
                    live.update(build_frame(devices, workers))
                except KeyboardInterrupt:
                    break
                except Exception as e:
                    live.update(Panel(f"hypervisor unreachable: {e}"))
                time.sleep(self.refresh_s)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--endpoint", default="http://127.0.0.1:8001")
    ap.add_argument("--once", action="store_true")
    args = ap.parse_args()
    if args.once:
        print(render_once(args.endpoint))
    else:
        HypervisorTUI(args.endpoint).run()


if __name__ == "__main__":
    main()
