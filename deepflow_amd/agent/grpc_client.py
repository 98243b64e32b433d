"""Minimal gRPC client for trident.Synchronizer/Sync (agent side).

Speaks the same HTTP/2 subset as control/grpc_server.py: preface +
SETTINGS, one request stream with literal-HPACK headers, gRPC-framed
protobuf DATA, and trailer-based status."""
from __future__ import annotations

import socket
import struct
from typing import Dict

from ..control.grpc_server import (F_DATA, F_HEADERS, F_SETTINGS,
                                   FLAG_END_HEADERS, FLAG_END_STREAM,
                                   FLAG_ACK, PREFACE, HpackDecoder,
                                   hpack_encode, frame, grpc_message)
from ..wire import pb, trident


def _read_frame(sock) -> tuple:
    hdr = b""
    while len(hdr) < 9:
        chunk = sock.recv(9 - len(hdr))
        if not chunk:
            raise ConnectionError("closed")
        hdr += chunk
    length = (hdr[0] << 16) | (hdr[1] << 8) | hdr[2]
    payload = b""
    while len(payload) < length:
        chunk = sock.recv(length - len(payload))
        if not chunk:
            raise ConnectionError("closed")
        payload += chunk
    sid = struct.unpack(">I", hdr[5:9])[0] & 0x7FFFFFFF
    return hdr[3], hdr[4], sid, payload


def grpc_sync(host: str, port: int, request: Dict,
              timeout: float = 5.0) -> Dict:
    """One Sync RPC; returns the decoded SyncResponse dict."""
    sock = socket.create_connection((host, port), timeout=timeout)
    dec = HpackDecoder()
    try:
        sock.sendall(PREFACE + frame(F_SETTINGS, 0, 0, b""))
        headers = hpack_encode([
            (":method", "POST"), (":scheme", "http"),
            (":path", "/trident.Synchronizer/Sync"),
            (":authority", f"{host}:{port}"),
            ("content-type", "application/grpc"), ("te", "trailers")])
        sock.sendall(frame(F_HEADERS, FLAG_END_HEADERS, 1, headers))
        body = grpc_message(pb.encode(request, trident.SYNC_REQUEST))
        sock.sendall(frame(F_DATA, FLAG_END_STREAM, 1, body))
        resp_data = b""
        grpc_status = None
        while True:
            ftype, flags, sid, payload = _read_frame(sock)
            if ftype == F_SETTINGS and not flags & FLAG_ACK:
                sock.sendall(frame(F_SETTINGS, FLAG_ACK, 0, b""))
            elif ftype == F_DATA and sid == 1:
                resp_data += payload
            elif ftype == F_HEADERS and sid == 1:
                hs = dict(dec.decode(payload))
                if "grpc-status" in hs:
                    grpc_status = int(hs["grpc-status"])
                if flags & FLAG_END_STREAM:
                    break
        if grpc_status not in (0, None):
            raise RuntimeError(f"grpc-status {grpc_status}")
        if len(resp_data) < 5:
            raise RuntimeError("empty gRPC response")
        mlen = struct.unpack(">I", resp_data[1:5])[0]
        return pb.decode(resp_data[5:5 + mlen], trident.SYNC_RESPONSE)
    finally:
        dec.close()
        sock.close()


def grpc_stream(host: str, port: int, path: str, request: Dict,
                request_schema, response_schema, max_msgs: int = 16,
                timeout: float = 10.0):
    """Server-streaming RPC (Push / Upgrade): yields decoded response
    messages until the server half-closes or max_msgs arrive."""
    sock = socket.create_connection((host, port), timeout=timeout)
    dec = HpackDecoder()
    got = 0
    try:
        sock.sendall(PREFACE + frame(F_SETTINGS, 0, 0, b""))
        headers = hpack_encode([
            (":method", "POST"), (":scheme", "http"),
            (":path", path),
            (":authority", f"{host}:{port}"),
            ("content-type", "application/grpc"), ("te", "trailers")])
        sock.sendall(frame(F_HEADERS, FLAG_END_HEADERS, 1, headers))
        body = grpc_message(pb.encode(request, request_schema))
        sock.sendall(frame(F_DATA, FLAG_END_STREAM, 1, body))
        buf = b""
        while got < max_msgs:
            ftype, flags, sid, payload = _read_frame(sock)
            if ftype == F_SETTINGS and not flags & FLAG_ACK:
                sock.sendall(frame(F_SETTINGS, FLAG_ACK, 0, b""))
            elif ftype == F_DATA and sid == 1:
                buf += payload
                while len(buf) >= 5:
                    mlen = struct.unpack(">I", buf[1:5])[0]
                    if len(buf) < 5 + mlen:
                        break
                    yield pb.decode(buf[5:5 + mlen], response_schema)
                    got += 1
                    buf = buf[5 + mlen:]
                    if got >= max_msgs:
                        return
            elif ftype == F_HEADERS and sid == 1 and flags & FLAG_END_STREAM:
                return
    finally:
        dec.close()
        sock.close()


def grpc_push(host: str, port: int, request: Dict, max_msgs: int = 4,
              timeout: float = 10.0):
    """trident.Synchronizer/Push: version-gated streamed SyncResponses."""
    return grpc_stream(host, port, "/trident.Synchronizer/Push", request,
                       trident.SYNC_REQUEST, trident.SYNC_RESPONSE,
                       max_msgs=max_msgs, timeout=timeout)


def grpc_upgrade(host: str, port: int, request: Dict,
                 timeout: float = 20.0) -> bytes:
    """trident.Synchronizer/Upgrade: reassemble the streamed package."""
    chunks = []
    for msg in grpc_stream(host, port, "/trident.Synchronizer/Upgrade",
                           request, trident.SYNC_REQUEST,
                           trident.UPGRADE_RESPONSE, max_msgs=1 << 20,
                           timeout=timeout):
        chunks.append(msg.get("content", b""))
    return b"".join(chunks)
