"""Alert policies: periodic DF-SQL evaluation -> alert_event rows.

The reference's alarm policies watch metrics and emit alert_event records
the UI lists; here a policy is (name, sql, column, op, threshold, level).
The evaluator runs each policy's query against the engine, compares the
chosen column per result row, and appends firing rows to the event
pipeline's alert_event table (queryable via `FROM alert_event`).
"""
from __future__ import annotations

import threading
from typing import Callable, Dict, List, Optional

_OPS: Dict[str, Callable] = {
    ">": lambda a, b: a > b, ">=": lambda a, b: a >= b,
    "<": lambda a, b: a < b, "<=": lambda a, b: a <= b,
    "=": lambda a, b: a == b, "!=": lambda a, b: a != b,
}

LEVEL_INFO, LEVEL_WARN, LEVEL_CRITICAL = 1, 2, 3


class AlertPolicy:
    def __init__(self, name: str, sql: str, column: str, op: str,
                 threshold: float, level: int = LEVEL_WARN,
                 target_column: Optional[str] = None):
        self.name = name
        self.sql = sql
        self.column = column
        self.op = op
        self.threshold = threshold
        self.level = level
        self.target_column = target_column
        self.fired = 0


class AlertEvaluator:
    def __init__(self, engine, event_pipeline, interval_s: float = 60.0):
        self.engine = engine
        self.events = event_pipeline
        self.interval_s = interval_s
        self.policies: List[AlertPolicy] = []
        self.evals = 0
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None

    def add_policy(self, policy: AlertPolicy) -> None:
        self.policies.append(policy)

    def evaluate_once(self) -> int:
        """Run every policy; returns alerts fired this pass."""
        fired = 0
        self.evals += 1
        for pol in self.policies:
            try:
                r = self.engine.query(pol.sql)
            except Exception:  # noqa: BLE001 — a broken policy must not
                continue       # kill the evaluator loop
            cols = r["columns"]
            if pol.column not in cols:
                continue
            ci = cols.index(pol.column)
            ti = cols.index(pol.target_column) \
                if pol.target_column in cols else None
            cmp = _OPS[pol.op]
            for row in r["values"]:
                v = row[ci]
                if v is None or not cmp(v, pol.threshold):
                    continue
                target = str(row[ti]) if ti is not None else ""
                self.events.add_alert_event(
                    pol.name, pol.level, target,
                    f"{pol.column}={v} {pol.op} {pol.threshold}")
                pol.fired += 1
                fired += 1
        return fired

    def start(self) -> None:
        def loop():
            while not self._stop.is_set():
                self.evaluate_once()
                self._stop.wait(self.interval_s)
        self._thread = threading.Thread(target=loop, daemon=True)
        self._thread.start()

    def stop(self) -> None:
        self._stop.set()
        if self._thread:
            self._thread.join(timeout=2)
