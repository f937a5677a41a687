"""Alert evaluator — SQL-template rules over the TSDB → alertmanager.

Reference: internal/alert/evaluator.go:27-186 (rules with query template,
threshold, interval, severity; firing/resolved transitions POSTed to an
alertmanager-compatible endpoint api.go:50).
"""
from __future__ import annotations

import json
import threading
import time
import urllib.request
from dataclasses import dataclass, field
from typing import Dict, List, Optional


@dataclass
class AlertRule:
    name: str
    query: str  # SQL over the TSDB points table; first column = value
    threshold: float
    op: str = ">"  # > | < | >= | <=
    interval_s: float = 60.0
    severity: str = "warning"
    summary: str = ""
    args: tuple = ()


@dataclass
class AlertState:
    rule: str
    firing: bool = False
    value: float = 0.0
    since: float = 0.0
    last_eval: float = 0.0


_OPS = {
    ">": lambda a, b: a > b,
    "<": lambda a, b: a < b,
    ">=": lambda a, b: a >= b,
    "<=": lambda a, b: a <= b,
}


def rules_from_config(raw_rules) -> List[AlertRule]:
    """Parse dynamic-config alert rules (config.py GlobalConfig
    .alert_rules — the reference's hot-reloaded alert config)."""

    out = []
    for r in raw_rules or []:
        try:
            out.append(AlertRule(
                name=r["name"],
                query=r["query"],
                threshold=float(r["threshold"]),
                op=r.get("op", ">"),
                interval_s=float(r.get("intervalSeconds", 60.0)),
                severity=r.get("severity", "warning"),
                summary=r.get("summary", "")))
        except (KeyError, TypeError, ValueError):
            continue
    return out


def default_rules() -> List[AlertRule]:
    """The reference ships similar defaults in its alert config."""

    return [
        AlertRule(
            name="PoolVramSaturation",
            query="SELECT avg(value) FROM points WHERE measurement="
                  "'tf_pool_metrics' AND field='allocated_vram' AND "
                  "ts_ns > (strftime('%s','now')-300)*1000000000",
            threshold=0.95 * 8 * 288 * 1024**3, op=">",
            severity="critical",
            summary="pool VRAM allocation above 95% of an 8-GPU node"),
        AlertRule(
            name="WorkerThrottledHigh",
            query="SELECT max(value) FROM points WHERE measurement="
                  "'tf_worker_metrics' AND field='throttled_ratio' AND "
                  "ts_ns > (strftime('%s','now')-300)*1000000000",
            threshold=0.5, op=">",
            summary="a worker is throttled >50% of the time"),
        AlertRule(
            name="SchedulerUnschedulable",
            query="SELECT max(value) FROM points WHERE measurement="
                  "'tf_system_metrics' AND field='unschedulable'",
            threshold=0, op=">",
            summary="pods cannot be scheduled"),
    ]


class AlertEvaluator:
    def __init__(self, tsdb, rules: Optional[List[AlertRule]] = None,
                 alertmanager_url: str = ""):
        self.tsdb = tsdb
        self.rules = rules if rules is not None else default_rules()
        self.alertmanager_url = alertmanager_url
        self.states: Dict[str, AlertState] = {}
        self.posted: List[dict] = []  # for tests / when no alertmanager
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None

    def evaluate(self, now: Optional[float] = None) -> List[AlertState]:
        """Evaluate due rules; returns states that changed."""

        now = now if now is not None else time.time()
        changed = []
        for rule in self.rules:
            st = self.states.setdefault(rule.name, AlertState(rule=rule.name))
            if now - st.last_eval < rule.interval_s and st.last_eval > 0:
                continue
            st.last_eval = now
            try:
                rows = self.tsdb.sql(rule.query, rule.args)
            except Exception:
                continue
            value = rows[0][0] if rows and rows[0] and rows[0][0] is not None \
                else None
            if value is None:
                continue
            st.value = float(value)
            firing = _OPS[rule.op](st.value, rule.threshold)
            if firing != st.firing:
                st.firing = firing
                st.since = now if firing else st.since
                changed.append(st)
                self._post(rule, st)
        return changed

    def _post(self, rule: AlertRule, st: AlertState):
        alert = {
            "labels": {"alertname": rule.name, "severity": rule.severity},
            "annotations": {"summary": rule.summary,
                            "value": str(st.value)},
            "status": "firing" if st.firing else "resolved",
        }
        self.posted.append(alert)
        if not self.alertmanager_url:
            return
        try:
            req = urllib.request.Request(
                f"{self.alertmanager_url}/api/v2/alerts",
                data=json.dumps([alert]).encode(),
                headers={"Content-Type": "application/json"})
            urllib.request.urlopen(req, timeout=5)
        except Exception:
            pass

    def start(self, interval_s: float = 15.0):
        self._stop.clear()

        def loop():
            while not self._stop.wait(interval_s):
                try:
                    self.evaluate()
                except Exception:
                    pass
        self._thread = threading.Thread(target=loop, daemon=True,
                                        name="alert-evaluator")
        self._thread.start()

    def stop(self):
        self._stop.set()
        if self._thread:
            self._thread.join(timeout=2)
