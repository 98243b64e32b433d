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


def test_shared_object_plugin(tmp_path):
    """Compile a real .so against the df_plugin.h ABI, dlopen it, and
    re-parse a custom-protocol session through it (reference
    plugin/shared_obj counterpart)."""
    import subprocess
    from deepflow_amd.agent.plugins import PluginHost
    from pathlib import Path
    hdr_dir = Path("deepflow_amd/agent/include").resolve()
    src = tmp_path / "toyproto.cpp"
    src.write_text(r'''
#include <cstring>
#include <cstdio>
#include "df_plugin.h"
extern "C" int df_plugin_parse(const uint8_t* p, uint32_t n,
                               uint16_t port, DfPluginInfo* out) {
    // TOY protocol: "TOY <verb> <key>\n"
    if (n < 5 || memcmp(p, "TOY ", 4) != 0) return 0;
    const char* sp = (const char*)memchr(p + 4, ' ', n - 4);
    if (!sp) return 0;
    size_t vl = sp - (const char*)(p + 4);
    if (vl >= sizeof out->req_type) vl = sizeof out->req_type - 1;
    memcpy(out->req_type, p + 4, vl);
    out->req_type[vl] = 0;
    snprintf(out->resource, sizeof out->resource, "%.*s",
             (int)(n - (sp - (const char*)p) - 1), sp + 1);
    char* nl = strchr(out->resource, '\n');
    if (nl) *nl = 0;
    snprintf(out->endpoint, sizeof out->endpoint, "toy:%u", port);
    out->status = 0;
    out->code = 200;
    return 1;
}
''')
    so = tmp_path / "toyproto.so"
    subprocess.run(["g++", "-shared", "-fPIC", f"-I{hdr_dir}", str(src),
                    "-o", str(so)], check=True)
    host = PluginHost()
    host.load_so(str(so))
    info = host.plugins[0](b"TOY GET user42\n", 9999)
    assert info is not None
    assert info.req_type == "GET" and info.resource == "user42"
    assert info.endpoint == "toy:9999" and info.code == 200
    # and through the record-rewrite path used by the agent
    from deepflow_amd.agent import Agent
    from deepflow_amd.agent.packets import eth_ipv4_tcp, PSH_ACK
    from deepflow_amd.wire import pb, flow_log, framing
    a = Agent(vtap_id=2)
    a.add_custom_protocol_port(9999)
    pkt = eth_ipv4_tcp(0x0A000001, 0x0A000002, 42000, 9999, seq=1,
                       flags=PSH_ACK, payload=b"TOY GET user42\n")
    a.packet(pkt, 10**18)
    rp = eth_ipv4_tcp(0x0A000002, 0x0A000001, 9999, 42000, seq=1,
                      flags=PSH_ACK, payload=b"OK\n")
    a.packet(rp, 10**18 + 10**6)
    a.tick(3 * 10**18)
    rewritten = host.process_l7_payload(a.drain(1))
    recs = [pb.decode(r, flow_log.APP_PROTO_LOGS_DATA)
            for r in framing.iter_records(rewritten)]
    assert any(d.get("req", {}).get("req_type") == "GET" and
               d["req"]["resource"] == "user42" for d in recs)
