"""Genesis: /proc scanner + controller inventory + GPID allocation."""
import os

import pytest
from fastapi.testclient import TestClient

from deepflow_amd.agent.proc_scanner import (scan_processes, scan_sockets,
                                             GenesisReporter)
from deepflow_amd.server import DeepflowServer


def test_scan_processes_self():
    procs = scan_processes()
    assert any(p["pid"] == os.getpid() for p in procs)
    me = next(p for p in procs if p["pid"] == os.getpid())
    assert "python" in me["name"] or "python" in me["cmdline"]


def test_scan_sockets():
    socks = scan_sockets()
    assert isinstance(socks, list)  # may be empty in minimal containers
    for s in socks[:5]:
        assert 0 <= s["local_port"] < 65536


def test_genesis_roundtrip():
    srv = DeepflowServer(device="cpu", tcp_port=0, segment_rows=1 << 9,
                         dict_capacity=1 << 10)
    client = TestClient(srv.app)

    def post(payload):
        return client.post("/v1/genesis/", json=payload).json()

    rep = GenesisReporter(agent_id=12, post_fn=post)
    gpids = rep.report()
    assert gpids[os.getpid()] > 0
    # stable across reports
    again = rep.report()
    assert again[os.getpid()] == gpids[os.getpid()]
    assert srv.controller.lookup_gpid(12, os.getpid()) == gpids[os.getpid()]
    inv = client.get("/v1/genesis/12").json()
    assert len(inv["processes"]) > 0
