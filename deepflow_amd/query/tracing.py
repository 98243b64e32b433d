"""Distributed-tracing assembly: spans -> trace tree.

The MI355X-native analog of the reference's tracing query path (the
SIGCOMM'23 join: spans related by trace_id, parent span ids,
syscall_trace_id and tcp_seq — SURVEY.md §2.5, libs/tracetree). Fetches the
trace's spans from the GPU store (string-hash filter on trace_id), links
parents three ways, and emits the tree.
"""
from __future__ import annotations

from typing import Dict, List, Optional


class DistributedTracer:
    def __init__(self, engine):
        self.engine = engine
        # assembled-tree summaries (reference: libs/tracetree rows written
        # back by the ingester; ours caches on assembly)
        self.tree_rows = []

    def fetch_spans(self, trace_id: str) -> List[Dict]:
        r = self.engine.query(
            "SELECT trace_id, span_id, parent_span_id, request_resource, "
            "service_name, start_time, end_time, response_status, "
            "syscall_trace_id_request, syscall_trace_id_response, "
            "req_tcp_seq, resp_tcp_seq, tap_side, flow_id "
            f"FROM l7_flow_log WHERE trace_id = '{trace_id}' LIMIT 10000")
        cols = r["columns"]
        return [dict(zip(cols, row)) for row in r["values"]]

    def assemble(self, trace_id: str) -> Dict:
        spans = self.fetch_spans(trace_id)
        by_span_id: Dict[str, int] = {}
        for i, s in enumerate(spans):
            sid = s.get("span_id")
            if sid:
                by_span_id.setdefault(sid, i)
        # syscall join: a span whose syscall_trace_id_request equals another
        # span's syscall_trace_id_response belongs below it (same thread
        # carried the request across the process)
        by_syscall_resp: Dict[int, int] = {}
        by_tcp_seq: Dict[int, int] = {}
        for i, s in enumerate(spans):
            v = s.get("syscall_trace_id_response") or 0
            if v:
                by_syscall_resp.setdefault(v, i)
            seq = s.get("req_tcp_seq") or 0
            if seq and s.get("tap_side") == 1:  # client-side emitter
                by_tcp_seq.setdefault(seq, i)
        nodes = []
        for i, s in enumerate(spans):
            parent: Optional[int] = None
            psid = s.get("parent_span_id")
            if psid and psid in by_span_id and by_span_id[psid] != i:
                parent = by_span_id[psid]
            if parent is None:
                v = s.get("syscall_trace_id_request") or 0
                if v and v in by_syscall_resp and by_syscall_resp[v] != i:
                    parent = by_syscall_resp[v]
            if parent is None and s.get("tap_side") != 1:
                seq = s.get("req_tcp_seq") or 0
                if seq and seq in by_tcp_seq and by_tcp_seq[seq] != i:
                    parent = by_tcp_seq[seq]
            nodes.append({
                "index": i,
                "span_id": s.get("span_id"),
                "parent_index": parent,
                "service": s.get("service_name"),
                "resource": s.get("request_resource"),
                "start_time": s.get("start_time"),
                "duration_ns": max((s.get("end_time") or 0) -
                                   (s.get("start_time") or 0), 0),
                "status": s.get("response_status"),
                "children": [],
            })
        roots = []
        for n in nodes:
            if n["parent_index"] is not None:
                nodes[n["parent_index"]]["children"].append(n["index"])
            else:
                roots.append(n["index"])
        result = {"trace_id": trace_id, "spans": nodes, "roots": roots,
                  "span_count": len(nodes)}
        if nodes:
            def depth(i, d=0):
                n = nodes[i]
                return max([d] + [depth(c, d + 1) for c in n["children"]])
            root = nodes[roots[0]] if roots else nodes[0]
            self.tree_rows.append({
                "time": (root["start_time"] or 0) // 10**9,
                "trace_id": trace_id,
                "span_count": len(nodes),
                "max_depth": max(depth(r) for r in roots) if roots else 0,
                "root_service": root["service"] or "",
                "duration_us": root["duration_ns"] // 1000,
            })
        return result

    def register(self, app) -> None:
        @app.get("/v1/tracing/{trace_id}")
        def tracing(trace_id: str):
            return self.assemble(trace_id)
