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
