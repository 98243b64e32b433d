

def test_agent_dfstats_to_server():
    """Agent self-metrics ride MSG_DFSTATS and land in deepflow_system."""
    from fastapi.testclient import TestClient
    from deepflow_amd.agent import Agent
    from deepflow_amd.agent.packets import http_session
    from deepflow_amd.server import DeepflowServer
    srv = DeepflowServer(device="cpu", tcp_port=0, segment_rows=1 << 10,
                         dict_capacity=1 << 12)
    srv.start()
    a = Agent(vtap_id=9, server=("127.0.0.1", srv.receiver.tcp_port))
    for frame, ts in http_session(0x0A000001, 0x0A000002, sport=40000,
                                  path="/x", code=200, t0=10**9):
        a.packet(frame, ts)
    n = a.flush_to_server(10**12)
    assert n >= 2  # data frame(s) + the dfstats frame
    import time
    deadline = time.time() + 10
    rows = []
    while time.time() < deadline:
        rows = [r for r in srv.system_rows
                if r.get("table") == "deepflow_agent"]
        if rows:
            break
        time.sleep(0.05)
    assert rows, "agent dfstats row not ingested"
    assert rows[0]["agent_id"] == "9"
    assert rows[0]["packets"] >= 4
    a.close()
    srv.stop()
