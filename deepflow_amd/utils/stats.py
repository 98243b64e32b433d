"""Self-telemetry counters (reference: server/libs/stats Countable loop).

Every pipeline stage registers a Counter; a StatsRegistry snapshots them
periodically and can ship them as dfstatsd Stats protobufs over the trident
framing (MESSAGE_TYPE_DFSTATS) back into the ingest path — the same
self-hosting telemetry loop the reference runs.
"""
from __future__ import annotations

import threading
import time
from typing import Dict, List, Optional


class Counter:
    def __init__(self, name: str, tags: Optional[Dict[str, str]] = None):
        self.name = name
        self.tags = tags or {}
        self._vals: Dict[str, float] = {}
        self._lock = threading.Lock()
        _default_registry.register(self)

    def add(self, key: str, v: float = 1) -> None:
        with self._lock:
            self._vals[key] = self._vals.get(key, 0) + v

    def set(self, key: str, v: float) -> None:
        with self._lock:
            self._vals[key] = v

    def snapshot(self, reset: bool = False) -> Dict[str, float]:
        with self._lock:
            snap = dict(self._vals)
            if reset:
                self._vals.clear()
        return snap


class StatsRegistry:
    def __init__(self):
        self._counters: List[Counter] = []
        self._lock = threading.Lock()

    def register(self, c: Counter) -> None:
        with self._lock:
            self._counters.append(c)

    def snapshot_all(self) -> List[dict]:
        now = int(time.time())
        out = []
        with self._lock:
            counters = list(self._counters)
        for c in counters:
            vals = c.snapshot()
            if not vals:
                continue
            out.append({
                "timestamp": now,
                "name": c.name,
                "tag_names": list(c.tags.keys()),
                "tag_values": list(c.tags.values()),
                "metrics_float_names": list(vals.keys()),
                "metrics_float_values": [float(v) for v in vals.values()],
            })
        return out

    def encode_dfstats(self) -> bytes:
        """Stats snapshots as a MSG_DFSTATS frame payload."""
        from ..wire import pb, metric, framing
        recs = [pb.encode(s, metric.STATS) for s in self.snapshot_all()]
        return framing.pack_records(recs)


_default_registry = StatsRegistry()


def default_registry() -> StatsRegistry:
    return _default_registry
