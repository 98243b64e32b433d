"""Agent integration collector: push endpoints -> trident relay -> server."""
import time

import pytest
from fastapi.testclient import TestClient

from deepflow_amd.agent.integration import IntegrationCollector
from deepflow_amd.server import DeepflowServer
from deepflow_amd.wire import pb
from tests.test_otel_control import mk_traces_data
from tests.test_prom_ingest import mk_write_request


@pytest.fixture(scope="module")
def stack():
    srv = DeepflowServer(device="cpu", tcp_port=0, segment_rows=1 << 10,
                         dict_capacity=1 << 12, time_base_s=1_700_000_000)
    srv.start()
    ic = IntegrationCollector(server=("127.0.0.1", srv.receiver.tcp_port),
                              agent_id=8)
    yield srv, TestClient(ic.app), ic
    srv.stop()


def _wait(cond, timeout=15):
    deadline = time.time() + timeout
    while time.time() < deadline:
        if cond():
            return True
        time.sleep(0.05)
    return False


def test_otel_push(stack):
    srv, client, ic = stack
    r = client.post("/api/v1/otel/trace", content=mk_traces_data())
    assert r.status_code == 200
    assert _wait(lambda: srv.l7.stats.spans_in >= 8)


def test_prometheus_push(stack):
    srv, client, ic = stack
    client.post("/api/v1/prometheus", content=mk_write_request())
    assert _wait(lambda: srv.prom.samples.n + len(srv.prom.samples._st_series) >= 20)


def test_profile_push(stack):
    srv, client, ic = stack
    client.post("/api/v1/profile/ingest",
                params={"name": "pyapp", "format": "folded"},
                content=b"main;slow_fn 9")
    assert _wait(lambda: len(srv.profiles.store.rows) >= 1)
    assert srv.profiles.store.id_to_loc[0] == b"main;slow_fn"
