"""In-kernel protocol inference spec — the single source of truth.

The reference maintains TWO hand-written copies of protocol inference:
the eBPF side (kernel/include/protocol_inference.h, 4.9k LoC C) and the
Rust packet side (each parser's check_payload). Here ONE declarative
table drives both consumers:

  - progs.py compiles it into BPF compare/branch instructions executed in
    the kernel on the first payload of each socket
  - tests cross-check `infer()` against the C++ packet parsers on golden
    payloads (tests/test_ebpf.py)

Matchers are first-bytes predicates only (what a kernel program can
evaluate in a handful of branches); protocols whose detection needs
ports or stateful reassembly (DNS over TCP, MQ binary frames) are left
to the userspace parser — the kernel hint is Optional, the userspace
parse is authoritative (same division as the reference: inference caches
a per-socket hint, protocol_logs re-parses).
"""
from __future__ import annotations

from typing import List, Optional, Tuple

from ..wire.const_enums import (
    L7_PROTOCOL_HTTP_1, L7_PROTOCOL_HTTP_2, L7_PROTOCOL_REDIS,
    L7_PROTOCOL_MYSQL, L7_PROTOCOL_POSTGRE, L7_PROTOCOL_KAFKA,
    L7_PROTOCOL_TLS, L7_PROTOCOL_MONGODB, L7_PROTOCOL_AMQP,
    L7_PROTOCOL_NATS, L7_PROTOCOL_MQTT,
)

# primitive checks (all ANDed within a rule):
#   ("prefix", b"GET ")          payload starts with bytes
#   ("byte_in", i, b"+-:$*")     payload[i] is one of the bytes
#   ("byte_eq", i, v)            payload[i] == v
#   ("byte_range", i, lo, hi)    lo <= payload[i] <= hi
#   ("min_len", n)               at least n bytes captured
#   ("u32be0_lenm4",)            BE u32 at 0 == syscall_len - 4 (kafka/framed)
Rule = Tuple

# (proto_id, name, [rule alternatives])
SPEC: List[Tuple[int, str, List[List[Rule]]]] = [
    (L7_PROTOCOL_HTTP_1, "http1", [
        [("prefix", b"GET ")], [("prefix", b"POST ")], [("prefix", b"PUT ")],
        [("prefix", b"HEAD ")], [("prefix", b"DELETE ")],
        [("prefix", b"OPTIONS ")], [("prefix", b"PATCH ")],
        [("prefix", b"HTTP/1.")],
    ]),
    (L7_PROTOCOL_HTTP_2, "http2", [
        [("prefix", b"PRI * HTTP/2")],
    ]),
    (L7_PROTOCOL_REDIS, "redis", [
        [("min_len", 4), ("byte_in", 0, b"*+-$:"),
         ("byte_range_or_digit", 1)],
    ]),
    (L7_PROTOCOL_TLS, "tls", [
        [("min_len", 6), ("byte_eq", 0, 0x16), ("byte_eq", 1, 0x03),
         ("byte_range", 2, 0x00, 0x04)],
    ]),
    (L7_PROTOCOL_POSTGRE, "postgresql", [
        # simple query: 'Q' + u32 len; common startup responses R/S/T/E
        [("min_len", 5), ("byte_in", 0, b"QPRSTE"), ("byte_eq", 1, 0)],
    ]),
    (L7_PROTOCOL_MYSQL, "mysql", [
        # [len3 LE][seq==0][cmd COM_QUERY(3)/COM_STMT_*]
        [("min_len", 5), ("byte_eq", 3, 0), ("byte_range", 4, 0x01, 0x1C),
         ("byte_eq", 2, 0)],
    ]),
    (L7_PROTOCOL_MONGODB, "mongodb", [
        # [msglen u32le][reqid][respto][opcode 2004/2010/2013 LE]
        [("min_len", 16), ("byte_eq", 12, 0xDD), ("byte_eq", 13, 0x07),
         ("byte_eq", 14, 0), ("byte_eq", 15, 0)],
    ]),
    (L7_PROTOCOL_AMQP, "amqp", [
        [("prefix", b"AMQP")],
    ]),
    (L7_PROTOCOL_NATS, "nats", [
        [("prefix", b"CONNECT {")], [("prefix", b"INFO {")],
        [("prefix", b"PUB ")], [("prefix", b"SUB ")], [("prefix", b"MSG ")],
    ]),
    (L7_PROTOCOL_MQTT, "mqtt", [
        # CONNECT fixed header 0x10 + 1-byte remaining-length, then the
        # protocol-name field \x00\x04MQTT at offset 2 (CONNECT-only form)
        [("min_len", 8), ("byte_eq", 0, 0x10), ("prefix_at", 2, b"\x00\x04MQTT")],
    ]),
    (L7_PROTOCOL_KAFKA, "kafka", [
        [("min_len", 12), ("u32be0_lenm4",), ("byte_eq", 4, 0),
         ("byte_range", 5, 0, 67), ("byte_eq", 6, 0),
         ("byte_range", 7, 0, 12)],
    ]),
]


def _check(rule: Rule, p: bytes, full_len: int) -> bool:
    kind = rule[0]
    if kind == "prefix":
        return p.startswith(rule[1])
    if kind == "prefix_at":
        off, pat = rule[1], rule[2]
        return p[off:off + len(pat)] == pat
    if kind == "byte_in":
        return len(p) > rule[1] and p[rule[1]] in rule[2]
    if kind == "byte_eq":
        return len(p) > rule[1] and p[rule[1]] == rule[2]
    if kind == "byte_range":
        return len(p) > rule[1] and rule[2] <= p[rule[1]] <= rule[3]
    if kind == "byte_range_or_digit":
        i = rule[1]
        return len(p) > i and (0x30 <= p[i] <= 0x39 or p[i] in b"+-OE")
    if kind == "min_len":
        return len(p) >= rule[1]
    if kind == "u32be0_lenm4":
        if len(p) < 4:
            return False
        return int.from_bytes(p[:4], "big") == full_len - 4
    raise ValueError(f"unknown rule {kind}")


def infer(payload: bytes, full_len: Optional[int] = None) -> int:
    """Python twin of the generated BPF matcher (the test oracle)."""
    full_len = len(payload) if full_len is None else full_len
    for proto, _name, alts in SPEC:
        for alt in alts:
            if all(_check(r, payload, full_len) for r in alt):
                return proto
    return 0
