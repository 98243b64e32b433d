"""Event pipelines: resource events, proc (perf) events, alert events.

Reference counterpart: server/ingester/event (decoder.go:309-330) — three
sources: controller resource-change events, agent ProcEvent streams
(IO/file-op/lifecycle), alert events. Rows are host-side (event volumes are
control-plane scale) and queryable as `event`, `perf_event`, `alert_event`.
"""
from __future__ import annotations

import time
from typing import Dict, List, Optional

from ..utils.stats import Counter
from ..wire import pb, framing

# ProcEvent wire schema (message/metric.proto:250-346 subset: IoEventData)
IO_EVENT_DATA = {
    1: ("bytes_count", 'u'),
    2: ("operation", 'u'),   # 0 write, 1 read
    3: ("latency", 'u'),
    4: ("filename", 'b'),
    9: ("file_type", 'u'),
}

PROC_EVENT = {
    1: ("pid", 'u'),
    2: ("thread_id", 'u'),
    3: ("coroutine_id", 'u'),
    4: ("process_kname", 'b'),
    5: ("start_time", 'u'),
    6: ("end_time", 'u'),
    7: ("event_type", 'u'),
    8: ("io_event_data", 'm', IO_EVENT_DATA),
    9: ("pod_id", 'u'),
}


class EventPipeline:
    def __init__(self, counter: Optional[Counter] = None):
        self.resource_events: List[Dict] = []
        self.perf_events: List[Dict] = []
        self.alert_events: List[Dict] = []
        self.counter = counter or Counter("ingester.event")

    # controller-originated resource changes (reference: recorder emits to
    # the ingester event queue)
    def add_resource_event(self, event_type: str, resource_type: str,
                           resource_id: int, resource_name: str = "",
                           description: str = "") -> None:
        self.resource_events.append({
            "time": int(time.time()),
            "event_type": event_type,          # create/delete/update...
            "resource_type": resource_type,    # pod/vm/service...
            "resource_id": resource_id,
            "resource_name": resource_name,
            "description": description,
        })
        self.counter.add("resource_events")

    def ingest_proc_events(self, payload: bytes) -> int:
        n = 0
        for rec in framing.iter_records(bytes(payload)):
            d = pb.decode(rec, PROC_EVENT)
            io = d.get("io_event_data", {})
            self.perf_events.append({
                "time": d.get("start_time", 0) // 1_000_000 or
                int(time.time()),
                "pid": d.get("pid", 0),
                "thread_id": d.get("thread_id", 0),
                "event_type": d.get("event_type", 0),
                "process_kname": d.get("process_kname", b"").decode(
                    "utf-8", "replace").rstrip("\0"),
                "pod_id": d.get("pod_id", 0),
                "bytes_count": io.get("bytes_count", 0),
                "operation": io.get("operation", 0),
                "latency": io.get("latency", 0),
                "filename": io.get("filename", b"").decode(
                    "utf-8", "replace").rstrip("\0"),
                "duration_us": max(d.get("end_time", 0) -
                                   d.get("start_time", 0), 0) // 1000,
            })
            n += 1
        self.counter.add("proc_events", n)
        return n

    def add_alert_event(self, policy_name: str, level: int,
                        target: str, message: str) -> None:
        self.alert_events.append({
            "time": int(time.time()),
            "policy_name": policy_name,
            "level": level,
            "target": target,
            "message": message,
        })
        self.counter.add("alert_events")
