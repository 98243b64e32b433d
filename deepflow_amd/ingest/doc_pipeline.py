"""Document (flow_metrics) ingest: agent Document pb stream -> rollup tables.

Reference counterpart: server/ingester/flow_metrics (unmarshaller ->
network/application tables). Agent-originated Documents land in their own
GPU-resident rollup tables (queryable as application.agent /
network.agent) so they compose with — rather than double-count — the
server-side span/flow rollups. Batch path: decode the payload's documents
into (key tuple, meter values) arrays, then one k_rollup_insert launch per
table (device mode) or an exact dict merge (cpu mode).
"""
from __future__ import annotations

from typing import Dict, List, Optional

from ..store.metrics import (APP_FIELDS, NET_FIELDS, RollupTable, TableDef,
                             CODE_VTAP, CODE_L3_EPC, CODE_SERVER_PORT,
                             CODE_PROTOCOL, CODE_L7_PROTOCOL, CODE_IP)
from ..utils.stats import Counter
from ..wire import pb, metric, framing

# agent-document tables: key = (time, vtap, epc, server_port, proto, ip)
_APP_TD = TableDef(
    "application.agent", "doc", 1,
    ("vtap_id", "l3_epc_id", "server_port", "l7_protocol", "ip"), "app",
    CODE_VTAP | CODE_L3_EPC | CODE_SERVER_PORT | CODE_L7_PROTOCOL | CODE_IP)
_NET_TD = TableDef(
    "network.agent", "doc", 1,
    ("vtap_id", "l3_epc_id", "server_port", "protocol", "ip"), "net",
    CODE_VTAP | CODE_L3_EPC | CODE_SERVER_PORT | CODE_PROTOCOL | CODE_IP)


class DocPipeline:
    def __init__(self, counter: Optional[Counter] = None,
                 device: str = "cpu", time_base_s: int = 0):
        self.time_base_s = time_base_s
        self.app_table = RollupTable(_APP_TD, time_base_s, device=device)
        self.net_table = RollupTable(_NET_TD, time_base_s, device=device)
        self.counter = counter or Counter("ingester.flow_metrics")

    # queried as application.agent / network.agent
    @property
    def app_rows(self) -> List[Dict]:
        return self.app_table.rows()

    @property
    def net_rows(self) -> List[Dict]:
        return self.net_table.rows()

    def ingest_payload(self, payload: bytes) -> int:
        n = 0
        app_k: List[tuple] = []
        app_v: List[List[int]] = []
        net_k: List[tuple] = []
        net_v: List[List[int]] = []
        for rec in framing.iter_records(bytes(payload)):
            d = pb.decode(rec, metric.DOCUMENT)
            self._handle(d, app_k, app_v, net_k, net_v)
            n += 1
        self.app_table.insert(app_k, app_v)
        self.net_table.insert(net_k, net_v)
        self.counter.add("docs_in", n)
        return n

    def _key(self, d: Dict, tag: Dict, proto_field: str) -> tuple:
        ip = tag.get("ip", b"")
        ip_int = int.from_bytes(ip[:4], "big") if ip else 0
        rel = max(d.get("timestamp", 0) - self.time_base_s, 0)
        return (rel, tag.get("vtap_id", 0),
                tag.get("l3_epc_id", 0) & 0xFFFFFFFF,
                tag.get("server_port", 0), tag.get(proto_field, 0), ip_int)

    def _handle(self, d: Dict, app_k, app_v, net_k, net_v) -> None:
        tag = d.get("tag", {}).get("field", {})
        meter = d.get("meter", {})
        app = meter.get("app")
        if app:
            traffic = app.get("traffic", {})
            lat = app.get("latency", {})
            anom = app.get("anomaly", {})
            app_k.append(self._key(d, tag, "l7_protocol"))
            app_v.append([traffic.get("request", 0),
                          traffic.get("response", 0),
                          anom.get("client_error", 0),
                          anom.get("server_error", 0),
                          lat.get("rrt_sum", 0),
                          lat.get("rrt_count", 0),
                          lat.get("rrt_max", 0)])
        flow = meter.get("flow")
        if flow:
            traffic = flow.get("traffic", {})
            lat = flow.get("latency", {})
            net_k.append(self._key(d, tag, "protocol"))
            net_v.append([traffic.get("byte_tx", 0), traffic.get("byte_rx", 0),
                          traffic.get("packet_tx", 0),
                          traffic.get("packet_rx", 0),
                          traffic.get("new_flow", 0),
                          traffic.get("closed_flow", 0),
                          lat.get("rtt_sum", 0), lat.get("rtt_count", 0),
                          lat.get("rtt_max", 0), 0])
        usage = meter.get("usage")
        if usage:
            net_k.append(self._key(d, tag, "protocol"))
            net_v.append([usage.get("byte_tx", 0), usage.get("byte_rx", 0),
                          usage.get("packet_tx", 0),
                          usage.get("packet_rx", 0), 0, 0, 0, 0, 0, 0])
