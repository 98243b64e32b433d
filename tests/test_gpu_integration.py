"""Full-stack GPU integration (-m gpu): C++ agent -> TCP -> receiver -> GPU
pipelines -> SQL / PromQL / Tempo / tracing on device tensors."""
import time

import pytest
import torch

pytestmark = pytest.mark.gpu

from fastapi.testclient import TestClient

from deepflow_amd.agent import Agent
from deepflow_amd.agent.packets import http_session, dns_session
from deepflow_amd.server import DeepflowServer

CLIENT, SERVER = 0x0A000001, 0x0A000002


def test_end_to_end_gpu():
    assert torch.cuda.is_available()
    srv = DeepflowServer(device="cuda", tcp_port=0, segment_rows=1 << 14,
                         dict_capacity=1 << 14, time_base_s=0)
    srv.start()
    try:
        a = Agent(vtap_id=3, server=("127.0.0.1", srv.receiver.tcp_port))
        a.add_cidr(0x0A000000, 8, epc=5)
        t0 = 10**9
        for i in range(50):
            for frame, ts in http_session(CLIENT + i, SERVER,
                                          sport=41000 + i,
                                          path=f"/api/item/{i % 5}",
                                          code=500 if i % 10 == 0 else 200,
                                          t0=t0 + i * 10**7):
                a.packet(frame, ts)
        for frame, ts in dns_session(CLIENT, SERVER):
            a.packet(frame, ts)
        a.flush_to_server(10**9 * 100)
        deadline = time.time() + 30
        while time.time() < deadline and srv.l7.stats.spans_in < 51:
            time.sleep(0.1)
        torch.cuda.synchronize()
        assert srv.l7.stats.spans_in == 51
        assert srv.l4.stats.flows_in == 51

        client = TestClient(srv.app)
        r = client.post("/v1/query/", json={
            "sql": "SELECT request_resource, Count(*) AS c FROM l7_flow_log "
                   "WHERE l7_protocol = 'HTTP' GROUP BY request_resource "
                   "ORDER BY c DESC"})
        vals = r.json()["result"]["values"]
        assert sum(v[1] for v in vals) == 50 and len(vals) == 5
        r2 = client.post("/v1/query/", json={
            "sql": "SELECT Count(*) AS c FROM l7_flow_log WHERE "
                   "response_code = 500"})
        assert r2.json()["result"]["values"] == [[5]]
        r3 = client.post("/v1/query/", json={
            "sql": "SELECT Sum(byte_tx) AS b FROM l4_flow_log"})
        assert r3.json()["result"]["values"][0][0] > 0
        # promql over GPU-built rollups
        r4 = client.get("/prom/api/v1/query", params={
            "query": "sum(increase(application_request[1h]))", "time": "3600"})
        assert float(r4.json()["data"]["result"][0]["value"][1]) == 51
        a.close()
    finally:
        srv.stop()
