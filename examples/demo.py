#!/usr/bin/env python3
"""End-to-end demo on one machine (CPU or GPU):

  python examples/demo.py

Starts the all-in-one server, drives it with the C++ agent engine (synthetic
HTTP/DNS/Redis traffic + an OTLP push + a profile push), then walks the
query surface: DF-SQL, PromQL, Tempo, trace tree, flame graph, exporter.
"""
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
from fastapi.testclient import TestClient

from deepflow_amd.agent import Agent
from deepflow_amd.agent.packets import http_session, dns_session, redis_session
from deepflow_amd.server import DeepflowServer
from deepflow_amd.wire import framing, pb, metric, flow_log

TRACE = "deadbeefcafe00112233445566778899"


def mk_traced_span(span_id, parent="", svc="front", res="/api/checkout",
                   t0=10**18, dur_ns=5_000_000):
    return {
        "base": {"start_time": t0, "end_time": t0 + dur_ns, "flow_id": 99,
                 "vtap_id": 1, "tap_side": 1,
                 "head": {"proto": 20, "msg_type": 2, "rrt": dur_ns // 1000},
                 "ip_src": 0x0A000001, "ip_dst": 0x0A000002,
                 "l3_epc_id_src": 3, "l3_epc_id_dst": 3,
                 "port_src": 40000, "port_dst": 8080, "protocol": 6},
        "req": {"req_type": "GET", "domain": "svc", "resource": res,
                "endpoint": res},
        "resp": {"status": 0, "code": 200},
        "trace_info": {"trace_id": TRACE, "span_id": span_id,
                       "parent_span_id": parent},
        "ext_info": {"service_name": svc},
    }


def main() -> None:
    device = "cuda" if torch.cuda.is_available() else "cpu"
    print(f"[demo] device={device}")
    srv = DeepflowServer(device=device, tcp_port=0, segment_rows=1 << 14,
                         dict_capacity=1 << 14, time_base_s=0)
    srv.start()
    client = TestClient(srv.app)

    # --- agent traffic ---------------------------------------------------
    a = Agent(vtap_id=1, server=("127.0.0.1", srv.receiver.tcp_port))
    a.add_cidr(0x0A000000, 8, epc=3)
    t0 = 10**9
    for i in range(200):
        for frame, ts in http_session(0x0A000001 + i % 20, 0x0A000002,
                                      sport=40000 + i,
                                      path=f"/api/v1/orders/{i % 8}",
                                      code=500 if i % 25 == 0 else 200,
                                      t0=t0 + i * 10**7):
            a.packet(frame, ts)
    for frame, ts in dns_session(0x0A000001, 0x0A000035):
        a.packet(frame, ts)
    for frame, ts in redis_session(0x0A000001, 0x0A000050):
        a.packet(frame, ts)
    a.flush_to_server(10**12, compress=True)
    deadline = time.time() + 20
    while time.time() < deadline and srv.l7.stats.spans_in < 202:
        time.sleep(0.05)
    print(f"[demo] ingested spans={srv.l7.stats.spans_in} "
          f"flows={srv.l4.stats.flows_in} "
          f"dict_entries={srv.l7.dict.n_entries()}")

    # --- profile push ----------------------------------------------------
    prof = {"name": "checkout", "format": "folded", "event_type": 1,
            "process_name": "checkout",
            "data": b"main;handler;db_query 80\nmain;handler;render 20",
            "timestamp": 1}
    srv.receiver.handle_frame(framing.encode_frame(
        framing.FrameHeader(msg_type=framing.MSG_PROFILE),
        framing.pack_records([pb.encode(prof, metric.PROFILE)])))

    # --- queries ----------------------------------------------------------
    def q(sql):
        r = client.post("/v1/query/", json={"sql": sql}).json()
        print(f"[sql] {sql}")
        print("      ", r["result"]["columns"])
        for row in r["result"]["values"][:5]:
            print("      ", row)

    q("SELECT l7_protocol, Count(*) AS c FROM l7_flow_log "
      "GROUP BY l7_protocol ORDER BY c DESC")
    q("SELECT request_resource, Count(*) AS c, "
      "Avg(response_duration) AS avg_us FROM l7_flow_log "
      "WHERE l7_protocol = 'HTTP' GROUP BY request_resource "
      "ORDER BY c DESC LIMIT 5")
    q("SELECT response_status, Percentile(response_duration, 95) AS p95 "
      "FROM l7_flow_log GROUP BY response_status")
    q("SELECT Sum(byte_tx) AS tx, Sum(byte_rx) AS rx FROM l4_flow_log")

    prom = client.get("/prom/api/v1/query", params={
        "query": "sum(rate(application_request[1m])) by (vtap_id)",
        "time": "30"}).json()
    print("[promql]", json.dumps(prom["data"]["result"])[:160])

    # --- distributed trace (multi-hop, pushed over the trident wire) -----
    spans = [mk_traced_span("s-root", svc="front"),
             mk_traced_span("s-mid", parent="s-root", svc="mid"),
             mk_traced_span("s-leaf", parent="s-mid", svc="back")]
    srv.receiver.handle_frame(framing.encode_frame(
        framing.FrameHeader(msg_type=framing.MSG_PROTOCOLLOG),
        framing.pack_records(
            [pb.encode(s, flow_log.APP_PROTO_LOGS_DATA) for s in spans])))
    tree = client.get(f"/v1/tracing/{TRACE}").json()
    chain = " -> ".join(n["service"] for n in tree["spans"])
    print(f"[trace] {TRACE}: {tree['span_count']} span(s): {chain}")

    flame = client.get("/v1/profile/flame",
                       params={"process_name": "checkout"}).json()
    print(f"[flame] total={flame['value']} "
          f"top={flame['children'][0]['name']}")

    blob = client.get("/v1/export/otlp", params={"limit": 50}).content
    print(f"[export] otlp bytes={len(blob)}")

    # --- alert policy over the ingested spans ---------------------------
    client.post("/v1/alert-policies/", json={
        "name": "5xx-spike", "column": "c", "op": ">=", "threshold": 1,
        "level": 3, "target_column": "request_resource",
        "sql": "SELECT request_resource, Count(*) AS c FROM l7_flow_log "
               "WHERE response_status = 'Server Error' "
               "GROUP BY request_resource"})
    fired = client.post("/v1/alert-policies/evaluate").json()["fired"]
    alerts = srv.engine.query(
        "SELECT policy_name, target, level FROM alert_event LIMIT 3")
    print(f"[alerts] fired={fired} first={alerts['values'][:1]}")

    # --- org isolation --------------------------------------------------
    from deepflow_amd.wire import flow_log as _fl
    org_span = dict(mk_traced_span("org9-span", svc="tenant"),
                    trace_info={})
    srv.receiver.handle_frame(framing.encode_frame(
        framing.FrameHeader(msg_type=framing.MSG_PROTOCOLLOG, org_id=9),
        framing.pack_records([pb.encode(org_span,
                                        _fl.APP_PROTO_LOGS_DATA)])))
    c9 = client.post("/v1/query/", headers={"X-Org-Id": "9"},
                     json={"sql": "SELECT Count(*) AS c FROM l7_flow_log"})
    print(f"[org 9] rows={c9.json()['result']['values'][0][0]} "
          f"(isolated from default org)")

    # --- UDP debug bus --------------------------------------------------
    from deepflow_amd.utils.debug_bus import debug_call
    st = debug_call(srv.debug_bus.port, "store")["result"]
    print(f"[debug-bus] l7_rows={st['l7_rows']} "
          f"dict_entries={st['dict_entries']}")
    print("[demo] done")
    a.close()
    srv.stop()


if __name__ == "__main__":
    main()
