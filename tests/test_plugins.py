"""Custom-protocol plugin tests (wasm-analog parse_payload surface)."""
from deepflow_amd.agent import Agent
from deepflow_amd.agent.packets import eth_ipv4_tcp, SYN, SYNACK, PSH_ACK
from deepflow_amd.agent.plugins import L7PluginInfo
from deepflow_amd.wire import pb, flow_log, framing

CLIENT, SERVER = 0x0A000001, 0x0A000002


def my_proto_parser(raw: bytes, port: int):
    # toy wire protocol: "MYP <verb> <key>"
    if not raw.startswith(b"MYP "):
        return None
    parts = raw.decode().split()
    return L7PluginInfo(req_type=parts[1], resource=parts[2],
                        endpoint=parts[1], domain="myproto",
                        status=0, code=0,
                        attributes={"myp.version": "1"})


def test_custom_protocol_plugin():
    a = Agent(vtap_id=2)
    a.add_custom_protocol_port(9999)
    a.register_plugin(my_proto_parser)
    req = b"MYP FETCH user:42"
    resp = b"OK 1"
    t0 = 10**9
    pkts = [
        (eth_ipv4_tcp(CLIENT, SERVER, 40001, 9999, SYN, 1), t0),
        (eth_ipv4_tcp(SERVER, CLIENT, 9999, 40001, SYNACK, 2, 2), t0 + 10**6),
        (eth_ipv4_tcp(CLIENT, SERVER, 40001, 9999, PSH_ACK, 2, 3, req),
         t0 + 2 * 10**6),
        (eth_ipv4_tcp(SERVER, CLIENT, 9999, 40001, PSH_ACK, 3, 2 + len(req),
                      resp), t0 + 3 * 10**6),
    ]
    for frame, ts in pkts:
        a.packet(frame, ts)
    a.tick(10**9 * 100)
    recs = [pb.decode(r, flow_log.APP_PROTO_LOGS_DATA)
            for r in framing.iter_records(a.drain(1))]
    assert len(recs) == 1
    r = recs[0]
    assert r["base"]["head"]["proto"] == 127
    assert r["req"]["req_type"] == "FETCH"
    assert r["req"]["resource"] == "user:42"
    assert r["req"]["domain"] == "myproto"
    assert "myp.version" in r["ext_info"]["attribute_names"]
    assert a.plugin_host.parsed == 1
    a.close()
