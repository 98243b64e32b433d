"""Replay the reference's golden pcap fixtures through the C++ agent engine
(reads fixtures from the read-only reference mount when present; skipped on
machines without it, e.g. GPU boxes). This validates our parsers against
real captures, the reference's own test corpus
(agent/resources/test/flow_generator)."""
import os
import struct

import pytest

from deepflow_amd.agent import Agent
from deepflow_amd.wire import pb, flow_log, framing

FIX = "/root/reference/agent/resources/test/flow_generator"

pytestmark = pytest.mark.skipif(not os.path.isdir(FIX),
                                reason="reference fixtures not mounted")


def read_pcap(path):
    """Minimal pcap reader -> [(ts_ns, frame)] supporting us/ns magic,
    both endiannesses, and linktype ethernet/linux-sll(113)."""
    data = open(path, "rb").read()
    magic = struct.unpack("<I", data[:4])[0]
    if magic == 0xA1B2C3D4:
        endian, scale = "<", 1000
    elif magic == 0xA1B23C4D:
        endian, scale = "<", 1
    elif magic == 0xD4C3B2A1:
        endian, scale = ">", 1000
    else:
        raise ValueError("bad pcap magic")
    linktype = struct.unpack(endian + "I", data[20:24])[0]
    pos = 24
    out = []
    while pos + 16 <= len(data):
        ts_s, ts_frac, incl, orig = struct.unpack(endian + "IIII",
                                                  data[pos:pos + 16])
        pos += 16
        frame = data[pos:pos + incl]
        pos += incl
        if linktype == 113:  # LINUX_SLL -> fake ethernet
            # sll: 2 pkttype, 2 arphrd, 2 addrlen, 8 addr, 2 proto
            proto = frame[14:16]
            frame = b"\x02\xbb\x00\x00\x00\x01\x02\xaa\x00\x00\x00\x01" + \
                proto + frame[16:]
        out.append((ts_s * 10**9 + ts_frac * scale, frame))
    return out


def replay(path):
    a = Agent(vtap_id=1)
    for ts, frame in read_pcap(path):
        a.packet(frame, ts)
    a.tick(1 << 62)
    l7 = [pb.decode(r, flow_log.APP_PROTO_LOGS_DATA)
          for r in framing.iter_records(a.drain(1))]
    l4 = [pb.decode(r, flow_log.TAGGED_FLOW)
          for r in framing.iter_records(a.drain(0))]
    stats = a.stats()
    a.close()
    return l7, l4, stats


def test_dns_pcap():
    l7, l4, stats = replay(f"{FIX}/dns/a-and-ns.pcap")
    assert stats["parse_errors"] == 0
    assert len(l7) >= 1
    assert all(r["base"]["head"]["proto"] == 120 for r in l7)
    domains = {r["req"]["domain"] for r in l7}
    assert any("." in d for d in domains), domains


def test_http_pcap():
    l7, l4, stats = replay(f"{FIX}/http/client-ip.pcap")
    assert len(l7) == 1
    r = l7[0]
    assert r["base"]["head"]["proto"] == 20
    assert r["req"]["req_type"] == "POST"
    assert r["req"]["resource"].startswith("/biz-inquiry-bff/")
    assert r["resp"]["code"] == 200


def test_redis_pcap():
    l7, l4, stats = replay(f"{FIX}/redis/redis.pcap")
    assert len(l7) >= 1
    assert all(r["base"]["head"]["proto"] == 80 for r in l7)
    cmds = {r["req"]["req_type"] for r in l7}
    assert cmds, cmds


def test_mysql_pcap():
    path = f"{FIX}/mysql"
    caps = [f for f in os.listdir(path) if f.endswith(".pcap")]
    hits = 0
    for c in caps[:6]:
        l7, _, _ = replay(f"{path}/{c}")
        hits += sum(1 for r in l7 if r["base"]["head"]["proto"] == 60)
    assert hits >= 1


def test_l4_flow_metrics_from_pcap():
    l7, l4, stats = replay(f"{FIX}/redis/redis.pcap")
    assert len(l4) >= 1
    f = l4[0]["flow"]
    assert f["metrics_peer_src"]["packet_count"] > 0
    assert f["flow_key"]["port_dst"] == 6379 or \
        f["flow_key"]["port_src"] == 6379
