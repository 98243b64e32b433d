"""Synthetic packet builder for agent tests (golden-fixture analog of the
reference's pcap test corpus, agent/resources/test/flow_generator)."""
from __future__ import annotations

import struct
from typing import List, Tuple


def eth_ipv4_tcp(src_ip: int, dst_ip: int, sport: int, dport: int,
                 flags: int, seq: int = 0, ack: int = 0,
                 payload: bytes = b"",
                 mac_src: int = 0x02AA00000001,
                 mac_dst: int = 0x02BB00000001) -> bytes:
    tcp_len = 20 + len(payload)
    tot = 20 + tcp_len
    eth = mac_dst.to_bytes(6, "big") + mac_src.to_bytes(6, "big") + b"\x08\x00"
    ip = struct.pack(">BBHHHBBH", 0x45, 0, tot, 0, 0, 64, 6, 0) + \
        src_ip.to_bytes(4, "big") + dst_ip.to_bytes(4, "big")
    tcp = struct.pack(">HHIIBBHHH", sport, dport, seq, ack, 0x50, flags,
                      65535, 0, 0)
    return eth + ip + tcp + payload


def eth_ipv4_udp(src_ip: int, dst_ip: int, sport: int, dport: int,
                 payload: bytes = b"") -> bytes:
    udp_len = 8 + len(payload)
    tot = 20 + udp_len
    eth = (0x02BB00000001).to_bytes(6, "big") + \
        (0x02AA00000001).to_bytes(6, "big") + b"\x08\x00"
    ip = struct.pack(">BBHHHBBH", 0x45, 0, tot, 0, 0, 64, 17, 0) + \
        src_ip.to_bytes(4, "big") + dst_ip.to_bytes(4, "big")
    udp = struct.pack(">HHHH", sport, dport, udp_len, 0)
    return eth + ip + udp + payload


SYN, SYNACK, ACK, PSH_ACK, FIN_ACK, RST = (0x02, 0x12, 0x10, 0x18, 0x11, 0x04)


def http_session(client_ip: int, server_ip: int, sport: int = 43210,
                 dport: int = 8080, path: str = "/api/x",
                 host: str = "svc.example.com", code: int = 200,
                 t0: int = 10**9) -> List[Tuple[bytes, int]]:
    """Full TCP handshake + HTTP request/response + close. Returns
    [(frame, ts_ns)]."""
    req = (f"GET {path} HTTP/1.1\r\nHost: {host}\r\n"
           f"User-Agent: test\r\n\r\n").encode()
    resp = (f"HTTP/1.1 {code} OK\r\nContent-Length: 5\r\n\r\nhello").encode()
    pkts = [
        (eth_ipv4_tcp(client_ip, server_ip, sport, dport, SYN, 1000), t0),
        (eth_ipv4_tcp(server_ip, client_ip, dport, sport, SYNACK, 5000,
                      1001), t0 + 2_000_000),
        (eth_ipv4_tcp(client_ip, server_ip, sport, dport, ACK, 1001, 5001),
         t0 + 4_000_000),
        (eth_ipv4_tcp(client_ip, server_ip, sport, dport, PSH_ACK, 1001,
                      5001, req), t0 + 5_000_000),
        (eth_ipv4_tcp(server_ip, client_ip, dport, sport, PSH_ACK, 5001,
                      1001 + len(req), resp), t0 + 9_000_000),
        (eth_ipv4_tcp(client_ip, server_ip, sport, dport, FIN_ACK,
                      1001 + len(req), 5001 + len(resp)), t0 + 11_000_000),
        (eth_ipv4_tcp(server_ip, client_ip, dport, sport, FIN_ACK,
                      5001 + len(resp), 1002 + len(req)), t0 + 12_000_000),
    ]
    return pkts


def dns_session(client_ip: int, server_ip: int, qname: str = "example.com",
                sport: int = 53535, t0: int = 10**9, rcode: int = 0):
    def encode_qname(name: str) -> bytes:
        out = b""
        for part in name.split("."):
            out += bytes([len(part)]) + part.encode()
        return out + b"\x00"

    q = struct.pack(">HHHHHH", 0x1234, 0x0100, 1, 0, 0, 0) + \
        encode_qname(qname) + struct.pack(">HH", 1, 1)
    r = struct.pack(">HHHHHH", 0x1234, 0x8180 | rcode, 1, 1, 0, 0) + \
        encode_qname(qname) + struct.pack(">HH", 1, 1)
    return [
        (eth_ipv4_udp(client_ip, server_ip, sport, 53, q), t0),
        (eth_ipv4_udp(server_ip, client_ip, 53, sport, r), t0 + 3_000_000),
    ]


def redis_session(client_ip: int, server_ip: int, sport: int = 41000,
                  t0: int = 10**9):
    req = b"*2\r\n$3\r\nGET\r\n$5\r\nmykey\r\n"
    resp = b"$5\r\nhello\r\n"
    return [
        (eth_ipv4_tcp(client_ip, server_ip, sport, 6379, SYN, 1), t0),
        (eth_ipv4_tcp(server_ip, client_ip, 6379, sport, SYNACK, 2, 2),
         t0 + 1_000_000),
        (eth_ipv4_tcp(client_ip, server_ip, sport, 6379, PSH_ACK, 2, 3,
                      req), t0 + 2_000_000),
        (eth_ipv4_tcp(server_ip, client_ip, 6379, sport, PSH_ACK, 3,
                      2 + len(req), resp), t0 + 4_000_000),
    ]
