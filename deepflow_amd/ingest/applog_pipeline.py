"""Application log ingest (agent logs / syslog / OTLP logs -> queryable
rows with dict-encoded fields). Reference: server/ingester/app_log.

Wire: MSG_APPLICATION_LOG / MSG_SYSLOG / MSG_AGENT_LOG frames carrying
line-oriented or record payloads; OTLP logs arrive via the OTLP route.
"""
from __future__ import annotations

import time
from typing import Dict, List, Optional

from ..utils.stats import Counter

SEVERITIES = {"debug": 7, "info": 6, "warn": 4, "warning": 4, "error": 3,
              "fatal": 2, "critical": 2}


class AppLogPipeline:
    def __init__(self, counter: Optional[Counter] = None):
        self.rows: List[Dict] = []
        # SmartEncoding for repetitive fields
        self.app_interner: Dict[str, int] = {}
        self.counter = counter or Counter("ingester.app_log")

    def _intern(self, s: str) -> int:
        i = self.app_interner.get(s)
        if i is None:
            i = len(self.app_interner)
            self.app_interner[s] = i
        return i

    def ingest_lines(self, payload: bytes, agent_id: int = 0,
                     log_type: str = "system") -> int:
        """Line format: '<ts> <severity> <app> <body...>' with graceful
        fallback for free-form lines."""
        n = 0
        for line in payload.splitlines():
            line = line.decode("utf-8", "replace").strip()
            if not line:
                continue
            parts = line.split(" ", 3)
            ts = int(time.time())
            sev = 6
            app = ""
            body = line
            if len(parts) >= 4 and parts[0].isdigit():
                ts = int(parts[0])
                sev = SEVERITIES.get(parts[1].lower(), 6)
                app = parts[2]
                body = parts[3]
            self.rows.append({
                "time": ts,
                "agent_id": agent_id,
                "log_type": log_type,
                "severity": sev,
                "app_service": app,
                "app_id": self._intern(app),
                "body": body,
            })
            n += 1
        self.counter.add("logs_in", n)
        return n

    def ingest_rows(self, rows: List[Dict]) -> int:
        """Append pre-converted rows (OTLP logs path)."""
        for r in rows:
            app = r.get("app_service", "")
            r.setdefault("app_id", self._intern(app))
            self.rows.append(r)
        self.counter.add("logs_in", len(rows))
        return len(rows)

    def search(self, substr: str = "", severity_max: int = 7,
               limit: int = 100) -> List[Dict]:
        out = []
        for r in reversed(self.rows):
            if r["severity"] <= severity_max and \
                    (not substr or substr in r["body"]):
                out.append(r)
                if len(out) >= limit:
                    break
        return out
