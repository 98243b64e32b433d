"""Minimal schema-driven protobuf (proto3) runtime.

This is the authoritative wire codec for the framework's protobuf ABI: the
message schemas in `flow_log.py` / `metric.py` mirror the reference's
`message/flow_log.proto` and `message/metric.proto` field numbers so that the
bytes we emit/ingest are byte-compatible with deepflow agents/servers
(reference: /root/reference/message/*.proto). Messages are plain dicts; a
schema maps field number -> (name, kind[, sub-schema]).

Kinds:
  'u'   unsigned varint (uint32/uint64/bool/enum)
  'i'   signed varint, two's-complement 64-bit (int32/int64)
  'd'   double (wire type 1, little-endian f64)
  'x'   fixed64 unsigned (wire type 1; OTLP time fields)
  's'   utf-8 string (wire type 2)
  'b'   bytes (wire type 2)
  'm'   nested message (wire type 2), third tuple element = sub-schema
Repeated fields: prefix kind with '*' (scalars encode packed, per proto3).

The hot ingest path does NOT use this module: GPU kernels parse the wire
format directly (ops/csrc/decode_l7.hip) and the C++ generator emits it
(ops/csrc/gen_cpu.cpp). This module exists for tests, golden fixtures and
low-rate control-plane messages.
"""
from __future__ import annotations

import struct
from typing import Any, Dict, Tuple

MASK64 = (1 << 64) - 1


def write_varint(out: bytearray, v: int) -> None:
    v &= MASK64
    while True:
        b = v & 0x7F
        v >>= 7
        if v:
            out.append(b | 0x80)
        else:
            out.append(b)
            return


def read_varint(buf: memoryview, pos: int) -> Tuple[int, int]:
    result = 0
    shift = 0
    while True:
        b = buf[pos]
        pos += 1
        result |= (b & 0x7F) << shift
        if not (b & 0x80):
            return result & MASK64, pos
        shift += 7
        if shift >= 70:
            raise ValueError("varint too long")


def _encode_field(out: bytearray, num: int, kind: str, val: Any, sub=None) -> None:
    if kind == 'u':
        if val == 0:
            return
        write_varint(out, (num << 3) | 0)
        write_varint(out, int(val))
    elif kind == 'i':
        if val == 0:
            return
        write_varint(out, (num << 3) | 0)
        write_varint(out, int(val) & MASK64)
    elif kind == 'd':
        if val == 0.0:
            return
        write_varint(out, (num << 3) | 1)
        out += struct.pack('<d', float(val))
    elif kind == 'x':
        if val == 0:
            return
        write_varint(out, (num << 3) | 1)
        out += struct.pack('<Q', int(val))
    elif kind == 's':
        data = val.encode('utf-8') if isinstance(val, str) else bytes(val)
        if not data:
            return
        write_varint(out, (num << 3) | 2)
        write_varint(out, len(data))
        out += data
    elif kind == 'b':
        if not val:
            return
        write_varint(out, (num << 3) | 2)
        write_varint(out, len(val))
        out += bytes(val)
    elif kind == 'm':
        if val is None:
            return
        body = encode(val, sub)
        write_varint(out, (num << 3) | 2)
        write_varint(out, len(body))
        out += body
    else:
        raise ValueError(f"bad kind {kind}")


def encode(msg: Dict[str, Any], schema: Dict[int, tuple]) -> bytes:
    """Encode dict -> proto3 bytes. Fields emitted in ascending field order."""
    out = bytearray()
    for num in sorted(schema):
        spec = schema[num]
        name, kind = spec[0], spec[1]
        sub = spec[2] if len(spec) > 2 else None
        if name not in msg:
            continue
        val = msg[name]
        if kind.startswith('*'):
            base = kind[1:]
            if val is None or len(val) == 0:
                continue
            if base in ('u', 'i', 'd'):
                # proto3 packed encoding
                body = bytearray()
                for v in val:
                    if base == 'd':
                        body += struct.pack('<d', float(v))
                    else:
                        write_varint(body, int(v) & MASK64)
                write_varint(out, (num << 3) | 2)
                write_varint(out, len(body))
                out += body
            else:
                for v in val:
                    _encode_field(out, num, base, v, sub)
        else:
            _encode_field(out, num, kind, val, sub)
    return bytes(out)


def _sign64(v: int) -> int:
    return v - (1 << 64) if v >= (1 << 63) else v


def decode(data, schema: Dict[int, tuple], *, _mv=None) -> Dict[str, Any]:
    """Decode proto3 bytes -> dict. Unknown fields are skipped."""
    buf = memoryview(data) if _mv is None else _mv
    pos, end = 0, len(buf)
    out: Dict[str, Any] = {}
    while pos < end:
        key, pos = read_varint(buf, pos)
        num, wt = key >> 3, key & 7
        spec = schema.get(num)
        if wt == 0:
            v, pos = read_varint(buf, pos)
            if spec:
                name, kind = spec[0], spec[1]
                base = kind.lstrip('*')
                val = _sign64(v) if base == 'i' else v
                if kind.startswith('*'):
                    out.setdefault(name, []).append(val)
                else:
                    out[name] = val
        elif wt == 1:
            raw = bytes(buf[pos:pos + 8])
            pos += 8
            if spec:
                name, kind = spec[0], spec[1]
                base = kind.lstrip('*')
                if base == 'x':
                    val = struct.unpack('<Q', raw)[0]
                else:
                    val = struct.unpack('<d', raw)[0]
                if kind.startswith('*'):
                    out.setdefault(name, []).append(val)
                else:
                    out[name] = val
        elif wt == 2:
            ln, pos = read_varint(buf, pos)
            chunk = buf[pos:pos + ln]
            pos += ln
            if spec:
                name, kind = spec[0], spec[1]
                sub = spec[2] if len(spec) > 2 else None
                rep = kind.startswith('*')
                base = kind.lstrip('*')
                if base == 's':
                    val: Any = bytes(chunk).decode('utf-8', 'replace')
                elif base == 'b':
                    val = bytes(chunk)
                elif base == 'm':
                    val = decode(None, sub, _mv=chunk)
                elif base in ('u', 'i', 'd') and rep:
                    # packed repeated scalars
                    vals = []
                    p2 = 0
                    while p2 < len(chunk):
                        if base == 'd':
                            vals.append(struct.unpack('<d', bytes(chunk[p2:p2 + 8]))[0])
                            p2 += 8
                        else:
                            v, p2 = read_varint(chunk, p2)
                            vals.append(_sign64(v) if base == 'i' else v)
                    out.setdefault(name, []).extend(vals)
                    continue
                else:
                    raise ValueError(f"field {num}: wire type 2 for kind {kind}")
                if rep:
                    out.setdefault(name, []).append(val)
                else:
                    out[name] = val
        elif wt == 5:
            pos += 4
        else:
            raise ValueError(f"unsupported wire type {wt}")
    return out
