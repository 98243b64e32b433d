"""Kafka sink: enriched rows -> a Kafka topic over the Produce protocol.

Speaks Produce v2 with MessageSet v1 framing directly over TCP (the same
wire our agent-side Kafka parser understands) — no client library in the
image, none needed. Reference counterpart: the ingester's Kafka exporter
(server/ingester/exporters/kafka).
"""
from __future__ import annotations

import json
import socket
import struct
import zlib
from typing import Dict, List, Optional, Tuple


def _str(s: str) -> bytes:
    b = s.encode()
    return struct.pack(">h", len(b)) + b


def _bytes(b: Optional[bytes]) -> bytes:
    if b is None:
        return struct.pack(">i", -1)
    return struct.pack(">i", len(b)) + b


def _message_v1(value: bytes, key: Optional[bytes] = None,
                timestamp_ms: int = 0) -> bytes:
    body = (bytes([1, 0]) +                      # magic=1, attributes=0
            struct.pack(">q", timestamp_ms) +
            _bytes(key) + _bytes(value))
    crc = zlib.crc32(body) & 0xFFFFFFFF
    msg = struct.pack(">I", crc) + body
    return struct.pack(">q", 0) + struct.pack(">i", len(msg)) + msg


def encode_produce_v2(topic: str, messages: List[bytes],
                      correlation_id: int = 1,
                      client_id: str = "deepflow-amd",
                      acks: int = 1, timeout_ms: int = 5000,
                      partition: int = 0) -> bytes:
    record_set = b"".join(messages)
    body = (struct.pack(">hh", 0, 2) +           # api_key=Produce, v2
            struct.pack(">i", correlation_id) +
            _str(client_id) +
            struct.pack(">hi", acks, timeout_ms) +
            struct.pack(">i", 1) + _str(topic) +
            struct.pack(">i", 1) + struct.pack(">i", partition) +
            struct.pack(">i", len(record_set)) + record_set)
    return struct.pack(">i", len(body)) + body


def decode_produce_response_v2(data: bytes) -> Dict:
    pos = 4  # skip length
    corr = struct.unpack_from(">i", data, pos)[0]
    pos += 4
    (n_topics,) = struct.unpack_from(">i", data, pos)
    pos += 4
    out = {"correlation_id": corr, "topics": {}}
    for _ in range(n_topics):
        (tl,) = struct.unpack_from(">h", data, pos)
        pos += 2
        topic = data[pos:pos + tl].decode()
        pos += tl
        (n_parts,) = struct.unpack_from(">i", data, pos)
        pos += 4
        parts = {}
        for _ in range(n_parts):
            part, err, offset = struct.unpack_from(">ihq", data, pos)
            pos += 14
            # v2 adds log_append_time
            pos += 8
            parts[part] = {"error": err, "offset": offset}
        out["topics"][topic] = parts
    return out


class KafkaExporter:
    """Batches rows as JSON messages into one topic."""

    def __init__(self, brokers: List[Tuple[str, int]], topic: str,
                 client_id: str = "deepflow-amd"):
        self.brokers = list(brokers)
        self.topic = topic
        self.client_id = client_id
        self._sock: Optional[socket.socket] = None
        self._corr = 0
        self.sent = 0
        self.errors = 0

    def _connect(self) -> socket.socket:
        if self._sock is not None:
            return self._sock
        last = None
        for host, port in self.brokers:  # first healthy broker wins
            try:
                self._sock = socket.create_connection((host, port),
                                                      timeout=5)
                return self._sock
            except OSError as e:
                last = e
        raise ConnectionError(f"no broker reachable: {last}")

    def send_rows(self, rows: List[Dict], timestamp_ms: int = 0) -> Dict:
        msgs = [_message_v1(json.dumps(r, default=str).encode(),
                            timestamp_ms=timestamp_ms) for r in rows]
        self._corr += 1
        req = encode_produce_v2(self.topic, msgs,
                                correlation_id=self._corr,
                                client_id=self.client_id)
        try:
            sock = self._connect()
            sock.sendall(req)
            hdr = b""
            while len(hdr) < 4:
                hdr += sock.recv(4 - len(hdr))
            (ln,) = struct.unpack(">i", hdr)
            body = b""
            while len(body) < ln:
                body += sock.recv(ln - len(body))
            resp = decode_produce_response_v2(hdr + body)
        except OSError:
            self.errors += 1
            self._sock = None
            raise
        errs = [p["error"] for t in resp["topics"].values()
                for p in t.values() if p["error"]]
        if errs:
            self.errors += 1
        else:
            self.sent += len(rows)
        return resp

    def export_query(self, engine, sql: str) -> Dict:
        """Run a DF-SQL query and ship the rows."""
        r = engine.query(sql)
        rows = [dict(zip(r["columns"], v)) for v in r["values"]]
        return self.send_rows(rows)

    def close(self) -> None:
        if self._sock:
            self._sock.close()
            self._sock = None
