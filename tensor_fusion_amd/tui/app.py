"""Hypervisor terminal UI — the node operator's live dashboard.

Reference: pkg/hypervisor/tui/ + cmd/hypervisor-tui (a 2.4 kLoC
bubbletea app over the hypervisor HTTP API: device view, worker view,
metrics view, shm inspector dialog, key navigation). This is the
rich-based MI355X equivalent with the same surface:

  views    [1] devices   [2] workers   [3] metrics   [4] tier/pressure
  keys     1-4/tab switch view · j/k or arrows select row · enter opens
           the shm inspector for the selected worker · r refresh ·
           q quit · ? help overlay
  modes    --once prints a single frame (tests/headless); the live loop
           reads keys raw (termios cbreak + select, no extra deps)

Rendering is split into pure frame builders over plain dicts so tests
drive every view and the selection/dialog state machine without a
terminal or a hypervisor.
"""
from __future__ import annotations

import argparse
import json
import time
import urllib.request
from dataclasses import dataclass, field
from typing import Dict, List, Optional

from rich.console import Console, Group
from rich.panel import Panel
from rich.table import Table
from rich.text import Text

VIEWS = ("devices", "workers", "metrics", "tier")


def _fetch(base: str, path: str):
    with urllib.request.urlopen(f"{base}{path}", timeout=3) as r:
        return json.loads(r.read())


def _gib(n) -> str:
    return f"{(n or 0) / (1 << 30):.1f}"


def _bar(frac: float, width: int = 18) -> Text:
    frac = max(0.0, min(1.0, frac))
    filled = int(frac * width)
    color = "green" if frac < 0.7 else ("yellow" if frac < 0.9 else "red")
    t = Text("█" * filled, style=color)
    t.append("░" * (width - filled), style="grey37")
    t.append(f" {frac * 100:3.0f}%")
    return t


# ------------------------------------------------------------ frame state


@dataclass
class UiState:
    view: str = "devices"
    selected: int = 0
    show_help: bool = False
    shm_dialog: Optional[dict] = None  # worker record being inspected
    status: str = ""
    devices: List[dict] = field(default_factory=list)
    workers: List[dict] = field(default_factory=list)
    metrics: List[dict] = field(default_factory=list)
    tier: List[dict] = field(default_factory=list)

    def rows(self) -> List[dict]:
        return {"devices": self.devices, "workers": self.workers,
                "metrics": self.metrics, "tier": self.tier}[self.view]

    # ------------------------------------------------------- key machine

    def handle_key(self, key: str) -> bool:
        """Returns False when the app should exit."""

        if self.show_help:
            self.show_help = False
            return True
        if self.shm_dialog is not None:
            if key in ("q", "\x1b", "enter", "\n"):
                self.shm_dialog = None
            return True
        if key == "q":
            return False
        if key == "?":
            self.show_help = True
        elif key in ("1", "2", "3", "4"):
            self.view = VIEWS[int(key) - 1]
            self.selected = 0
        elif key == "\t":
            self.view = VIEWS[(VIEWS.index(self.view) + 1) % len(VIEWS)]
            self.selected = 0
        elif key in ("j", "down"):
            self.selected = min(self.selected + 1,
                                max(0, len(self.rows()) - 1))
        elif key in ("k", "up"):
            self.selected = max(0, self.selected - 1)
        elif key in ("enter", "\n", "\r") and self.view == "workers":
            rows = self.rows()
            if rows:
                self.shm_dialog = rows[min(self.selected, len(rows) - 1)]
        return True


# ------------------------------------------------------------ view frames


def devices_frame(devices: List[dict], selected: int = -1) -> Panel:
    t = Table(title="Devices (MI355X)", expand=True)
    for col in ("idx", "uuid", "VRAM", "used/total GiB", "busy",
                "CUs", "partition", "workers"):
        t.add_column(col)
    for i, d in enumerate(devices):
        total = d.get("vram_total", 0) or 1
        style = "reverse" if i == selected else ""
        t.add_row(str(d.get("index", "?")), d.get("uuid", "")[:22],
                  _bar(d.get("vram_used", 0) / total),
                  f"{_gib(d.get('vram_used'))}/{_gib(total)}",
                  _bar(d.get("busy_percent", 0) / 100.0, 10),
                  str(d.get("compute_units", 256)),
                  d.get("compute_partition", "SPX"),
                  str(d.get("worker_count", 0)), style=style)
    return Panel(t, title="[1] devices")


def workers_frame(workers: List[dict], selected: int = -1) -> Panel:
    t = Table(title="Workers (vGPUs)", expand=True)
    for col in ("pod", "qos", "iso", "VRAM used/limit", "comp %",
                "ERL rate", "tokens", "throttled ms", "hb age"):
        t.add_column(col)
    now = time.time()
    for i, w in enumerate(workers):
        lim = w.get("limits", {})
        usage = w.get("usage", {})
        hb = w.get("heartbeat_ts", 0)
        style = "reverse" if i == selected else ""
        t.add_row(
            f"{w.get('namespace', '')}/{w.get('pod', '')}"[:30],
            w.get("qos", ""), w.get("isolation", "")[:4],
            f"{_gib(usage.get('vram'))}/{_gib(lim.get('vram'))} GiB",
            f"{lim.get('compute_percent', 100):.0f}",
            f"{usage.get('erl_rate', 0):.0f}",
            f"{usage.get('erl_tokens', 0):.0f}",
            f"{usage.get('block_ns', 0) / 1e6:.0f}",
            f"{max(0, now - hb):.0f}s" if hb else "-", style=style)
    return Panel(t, title="[2] workers  (enter → shm inspector)")


def metrics_frame(metrics: List[dict], selected: int = -1) -> Panel:
    t = Table(title="Node metrics (60s window)", expand=True)
    for col in ("metric", "value", "per-worker breakdown"):
        t.add_column(col)
    for i, m in enumerate(metrics):
        style = "reverse" if i == selected else ""
        t.add_row(m.get("name", ""), str(m.get("value", "")),
                  m.get("detail", ""), style=style)
    return Panel(t, title="[3] metrics")


def tier_frame(tier: List[dict], selected: int = -1) -> Panel:
    t = Table(title="VRAM tier / pressure", expand=True)
    for col in ("worker", "HBM budget GiB", "resident", "host tier GiB",
                "demoted GiB", "promoted GiB", "pressure"):
        t.add_column(col)
    for i, w in enumerate(tier):
        style = "reverse" if i == selected else ""
        budget = w.get("budget", 0) or 1
        t.add_row(w.get("pod", "")[:30], _gib(budget),
                  _bar(w.get("resident", 0) / budget, 12),
                  _gib(w.get("host_bytes")), _gib(w.get("demoted")),
                  _gib(w.get("promoted")),
                  "[red]YES[/red]" if w.get("pressured") else "no",
                  style=style)
    return Panel(t, title="[4] tier/pressure")


def shm_dialog_frame(worker: dict) -> Panel:
    """The shm inspector (reference shm_dialog.go): raw per-device
    entries of the worker's limiter page."""

    t = Table(title=f"shm — {worker.get('namespace', '')}/"
                    f"{worker.get('pod', '')}", expand=True)
    for col in ("field", "value"):
        t.add_column(col)
    lim = worker.get("limits", {})
    usage = worker.get("usage", {})
    rows = [
        ("shm path", worker.get("shm_path", "")),
        ("uuid[0]", worker.get("device_uuid", "")),
        ("up_limit_percent", lim.get("compute_percent", "")),
        ("mem_limit_bytes", lim.get("vram", "")),
        ("pod_memory_used", usage.get("vram", "")),
        ("vmm_bytes (worker heap)", usage.get("vmm_bytes", 0)),
        ("erl_refill_rate", usage.get("erl_rate", "")),
        ("erl_capacity", usage.get("erl_capacity", "")),
        ("erl_tokens", usage.get("erl_tokens", "")),
        ("launch_count", usage.get("launches", "")),
        ("block_ns_total", usage.get("block_ns", "")),
        ("pids", ", ".join(str(p) for p in worker.get("pids", []))),
        ("flags", worker.get("flags", 0)),
        ("heartbeat age s",
         f"{max(0, time.time() - worker.get('heartbeat_ts', 0)):.1f}"
         if worker.get("heartbeat_ts") else "-"),
    ]
    for k, v in rows:
        t.add_row(str(k), str(v))
    return Panel(t, title="shm inspector — q/esc closes", style="cyan")


HELP = """\
 1-4 / tab   switch view            j/k or ↑/↓   select row
 enter       shm inspector (workers view)
 r           refresh now            q            quit / close dialog
 ?           this help
"""


def build_frame(state: UiState) -> Group:
    if state.show_help:
        return Group(Panel(HELP, title="help — any key closes"))
    if state.shm_dialog is not None:
        return Group(shm_dialog_frame(state.shm_dialog))
    sel = state.selected
    frames = {
        "devices": devices_frame(state.devices, sel),
        "workers": workers_frame(state.workers, sel),
        "metrics": metrics_frame(state.metrics, sel),
        "tier": tier_frame(state.tier, sel),
    }
    tabs = Text()
    for v in VIEWS:
        tabs.append(f" {v} ",
                    style="reverse bold" if v == state.view else "dim")
    status = Text(state.status or
                  "1-4/tab views · j/k select · enter inspect · ? help · "
                  "q quit", style="dim")
    return Group(tabs, frames[state.view], status)


# ---------------------------------------------------------------- data


def refresh(state: UiState, base: str):
    try:
        state.devices = _fetch(base, "/api/v1/devices").get("data", [])
        state.workers = _fetch(base, "/api/v1/workers").get("data", [])
        state.status = ""
    except Exception as e:
        state.status = f"hypervisor unreachable: {e}"
        return
    try:
        state.metrics = _fetch(base, "/api/v1/metrics").get("data", [])
    except Exception:
        state.metrics = summarize_metrics(state.devices, state.workers)
    state.tier = [{
        "pod": f"{w.get('namespace', '')}/{w.get('pod', '')}",
        "budget": w.get("limits", {}).get("vram", 0),
        "resident": w.get("usage", {}).get("vram", 0),
        "host_bytes": w.get("usage", {}).get("host_tier_bytes", 0),
        "demoted": w.get("usage", {}).get("demoted_bytes", 0),
        "promoted": w.get("usage", {}).get("promoted_bytes", 0),
        "pressured": bool(w.get("flags", 0) & 2),
    } for w in state.workers]


def summarize_metrics(devices: List[dict], workers: List[dict]
                      ) -> List[dict]:
    """Fallback metric rows computed client-side."""

    total_vram = sum(d.get("vram_total", 0) for d in devices)
    used_vram = sum(d.get("vram_used", 0) for d in devices)
    launches = sum(w.get("usage", {}).get("launches", 0) for w in workers)
    blocked = sum(w.get("usage", {}).get("block_ns", 0) for w in workers)
    per_worker = ", ".join(
        f"{w.get('pod', '?')}:{w.get('usage', {}).get('launches', 0)}"
        for w in workers[:6])
    return [
        {"name": "devices", "value": len(devices), "detail": ""},
        {"name": "workers", "value": len(workers), "detail": ""},
        {"name": "vram used GiB",
         "value": f"{used_vram / (1 << 30):.1f}/"
                  f"{total_vram / (1 << 30):.0f}", "detail": ""},
        {"name": "launches", "value": launches, "detail": per_worker},
        {"name": "throttled ms", "value": f"{blocked / 1e6:.0f}",
         "detail": ""},
    ]


# ---------------------------------------------------------------- app


def render_once(base: str, view: str = "devices",
                console: Optional[Console] = None) -> str:
    console = console or Console(record=True, width=120)
    state = UiState(view=view)
    refresh(state, base)
    console.print(build_frame(state))
    return console.export_text() if console.record else ""


def _read_key(timeout_s: float) -> Optional[str]:
    """Raw single-key read with timeout (cbreak + select)."""

    import select
    import sys
    import termios
    import tty
    fd = sys.stdin.fileno()
    old = termios.tcgetattr(fd)
    try:
        tty.setcbreak(fd)
        r, _, _ = select.select([sys.stdin], [], [], timeout_s)
        if not r:
            return None
        ch = sys.stdin.read(1)
        if ch == "\x1b":  # arrow keys
            r, _, _ = select.select([sys.stdin], [], [], 0.05)
            if r:
                seq = sys.stdin.read(2)
                return {"[A": "up", "[B": "down"}.get(seq, "\x1b")
        return ch
    finally:
        termios.tcsetattr(fd, termios.TCSADRAIN, old)


class HypervisorTUI:
    def __init__(self, base: str = "http://127.0.0.1:8001",
                 refresh_s: float = 2.0):
        self.base = base
        self.refresh_s = refresh_s
        self.state = UiState()

    def run(self):
        from rich.live import Live
        console = Console()
        refresh(self.state, self.base)
        last = time.time()
        with Live(build_frame(self.state), console=console,
                  refresh_per_second=8, screen=True) as live:
            while True:
                key = _read_key(0.2)
                if key == "r":
                    refresh(self.state, self.base)
                elif key is not None:
                    if not self.state.handle_key(key):
                        return
                if time.time() - last > self.refresh_s:
                    refresh(self.state, self.base)
                    last = time.time()
                live.update(build_frame(self.state))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--endpoint", default="http://127.0.0.1:8001")
    ap.add_argument("--once", action="store_true")
    ap.add_argument("--view", default="devices", choices=VIEWS)
    args = ap.parse_args()
    if args.once:
        print(render_once(args.endpoint, view=args.view))
    else:
        HypervisorTUI(args.endpoint).run()


if __name__ == "__main__":
    main()
