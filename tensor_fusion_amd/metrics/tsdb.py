"""Embedded time-series store — the GreptimeDB stand-in.

Reference: the operator ships influx lines through Vector into GreptimeDB
and reads them back over the MySQL protocol (internal/metrics/connect.go,
migrate.go) to power the autoscaler percentile histograms and SQL alert
rules. Single-node deployments of this stack embed the store instead:
sqlite3 tables with the same measurement/tag/field model, ingesting the
recorder's influx lines directly, with TTL retention.
"""
from __future__ import annotations

import json
import re
import sqlite3
import threading
import time
from typing import Dict, List, Optional, Tuple

_LINE_RE = re.compile(r"^([^,\s]+)(?:,([^ ]*))? ([^ ]+) (\d+)$")


def _unesc(s: str) -> str:
    return s.replace("\\ ", " ").replace("\\,", ",").replace("\\=", "=")


def parse_influx_line(line: str) -> Optional[Tuple[str, Dict, Dict, int]]:
    m = _LINE_RE.match(line.strip())
    if not m:
        return None
    meas, tagstr, fieldstr, ts = m.groups()
    tags = {}
    if tagstr:
        for kv in re.split(r"(?<!\\),", tagstr):
            if "=" in kv:
                k, v = kv.split("=", 1)
                tags[_unesc(k)] = _unesc(v)
    fields = {}
    for kv in re.split(r"(?<!\\),", fieldstr):
        if "=" not in kv:
            continue
        k, v = kv.split("=", 1)
        if v.endswith("i"):
            fields[_unesc(k)] = int(v[:-1])
        else:
            try:
                fields[_unesc(k)] = float(v)
            except ValueError:
                fields[_unesc(k)] = v.strip('"')
    return meas, tags, fields, int(ts)


class TSDB:
    def __init__(self, path: str = ":memory:", ttl_s: float = 7 * 86400):
        self._conn = sqlite3.connect(path, check_same_thread=False)
        self._lock = threading.Lock()
        self.ttl_s = ttl_s
        self._conn.execute(
            "CREATE TABLE IF NOT EXISTS points ("
            " measurement TEXT, ts_ns INTEGER, tags TEXT, field TEXT,"
            " value REAL)")
        self._conn.execute(
            "CREATE INDEX IF NOT EXISTS idx_points ON points"
            " (measurement, field, ts_ns)")

    def ingest_lines(self, lines: List[str]) -> int:
        rows = []
        for line in lines:
            p = parse_influx_line(line)
            if p is None:
                continue
            meas, tags, fields, ts = p
            tj = json.dumps(tags, sort_keys=True)
            for f, v in fields.items():
                if isinstance(v, (int, float)):
                    rows.append((meas, ts, tj, f, float(v)))
        with self._lock:
            self._conn.executemany(
                "INSERT INTO points VALUES (?,?,?,?,?)", rows)
            self._conn.commit()
        return len(rows)

    def query(self, measurement: str, field: str,
              tags: Optional[Dict[str, str]] = None,
              since_s: Optional[float] = None) -> List[Tuple[int, float]]:
        """Return [(ts_ns, value)] matching the tag subset, time-ordered."""

        q = ("SELECT ts_ns, tags, value FROM points WHERE measurement=?"
             " AND field=?")
        args: list = [measurement, field]
        if since_s is not None:
            q += " AND ts_ns >= ?"
            args.append(int((time.time() - since_s) * 1e9))
        q += " ORDER BY ts_ns"
        out = []
        with self._lock:
            rows = self._conn.execute(q, args).fetchall()
        for ts, tj, v in rows:
            if tags:
                t = json.loads(tj)
                if any(t.get(k) != v2 for k, v2 in tags.items()):
                    continue
            out.append((ts, v))
        return out

    def sql(self, query: str, args: tuple = ()) -> List[tuple]:
        """Raw SQL over the points table (the alert evaluator's interface —
        reference alert rules are SQL templates)."""

        with self._lock:
            return self._conn.execute(query, args).fetchall()

    def vacuum_expired(self) -> int:
        cutoff = int((time.time() - self.ttl_s) * 1e9)
        with self._lock:
            cur = self._conn.execute(
                "DELETE FROM points WHERE ts_ns < ?", (cutoff,))
            self._conn.commit()
            return cur.rowcount
