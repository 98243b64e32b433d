"""Document (flow_metrics) ingest: agent Document pb stream -> rollup rows.

Reference counterpart: server/ingester/flow_metrics (unmarshaller ->
network/application tables). Agent-originated Documents land in their own
row store (queryable as application.agent / network.agent) so they compose
with — rather than double-count — the server-side span/flow rollups.
"""
from __future__ import annotations

from typing import Dict, List, Optional

from ..utils.stats import Counter
from ..wire import pb, metric, framing


class DocPipeline:
    def __init__(self, counter: Optional[Counter] = None):
        self.app_rows: List[Dict] = []
        self.net_rows: List[Dict] = []
        self.counter = counter or Counter("ingester.flow_metrics")

    def ingest_payload(self, payload: bytes) -> int:
        n = 0
        for rec in framing.iter_records(bytes(payload)):
            d = pb.decode(rec, metric.DOCUMENT)
            self._handle(d)
            n += 1
        self.counter.add("docs_in", n)
        return n

    def _handle(self, d: Dict) -> None:
        tag = d.get("tag", {}).get("field", {})
        meter = d.get("meter", {})
        ip = tag.get("ip", b"")
        base = {
            "time": d.get("timestamp", 0),
            "vtap_id": tag.get("vtap_id", 0),
            "l3_epc_id": tag.get("l3_epc_id", 0),
            "server_port": tag.get("server_port", 0),
            "protocol": tag.get("protocol", 0),
            "l7_protocol": tag.get("l7_protocol", 0),
            "ip": ".".join(str(b) for b in ip) if ip else "",
        }
        app = meter.get("app")
        if app:
            traffic = app.get("traffic", {})
            lat = app.get("latency", {})
            anom = app.get("anomaly", {})
            row = dict(base)
            row.update({
                "request": traffic.get("request", 0),
                "response": traffic.get("response", 0),
                "client_error": anom.get("client_error", 0),
                "server_error": anom.get("server_error", 0),
                "timeout": anom.get("timeout", 0),
                "rrt_sum": lat.get("rrt_sum", 0),
                "rrt_count": lat.get("rrt_count", 0),
                "rrt_max": lat.get("rrt_max", 0),
            })
            self.app_rows.append(row)
        flow = meter.get("flow")
        if flow:
            traffic = flow.get("traffic", {})
            lat = flow.get("latency", {})
            row = dict(base)
            row.update({
                "byte_tx": traffic.get("byte_tx", 0),
                "byte_rx": traffic.get("byte_rx", 0),
                "packet_tx": traffic.get("packet_tx", 0),
                "packet_rx": traffic.get("packet_rx", 0),
                "new_flow": traffic.get("new_flow", 0),
                "closed_flow": traffic.get("closed_flow", 0),
                "rtt_sum": lat.get("rtt_sum", 0),
                "rtt_count": lat.get("rtt_count", 0),
                "rtt_max": lat.get("rtt_max", 0),
            })
            self.net_rows.append(row)
        usage = meter.get("usage")
        if usage:
            row = dict(base)
            row.update({k: usage.get(k, 0) for k in
                        ("byte_tx", "byte_rx", "packet_tx", "packet_rx")})
            self.net_rows.append(row)
