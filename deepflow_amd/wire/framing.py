"""Trident framed-protobuf transport codec.

The agent->ingester data plane (reference ports 20033/20035) frames each batch
as a 19-byte header followed by a payload of length-prefixed protobuf records.
Layout verified against both ends of the reference:
  encode: agent/src/sender/uniform_sender.rs:109-147
  decode: server/libs/datatype/droplet-message.go:180-245

  [0..4)   frame_size  u32 BE   whole frame including this header
  [4]      msg_type    u8       MessageType enum (droplet-message.go:37-61)
  [5..7)   version     u16 LE   == 0x8000
  [7]      encoder     u8       0 = raw, 3 = zstd (whole-payload)
  [8..12)  team_id     u32 LE
  [12..14) org_id      u16 LE
  [14..16) reserved
  [16..18) agent_id    u16 LE
  [18]     reserved
  payload: repeated [u32 LE pb_len][pb bytes]   (SimpleEncoder WritePB framing,
           server/libs/codec/simple_codec.go:229-247)
"""
from __future__ import annotations

import struct
from dataclasses import dataclass
from typing import Iterable, Iterator, List, Tuple

HEADER_LEN = 19
VERSION = 0x8000

# MessageType enum values (droplet-message.go:37-61)
MSG_COMPRESS = 0
MSG_SYSLOG = 1
MSG_SERVER_DFSTATS = 2
MSG_METRICS = 3
MSG_TAGGEDFLOW = 4
MSG_PROTOCOLLOG = 5
MSG_OPENTELEMETRY = 6
MSG_PROMETHEUS = 7
MSG_TELEGRAF = 8
MSG_PACKETSEQUENCE = 9
MSG_DFSTATS = 10
MSG_OPENTELEMETRY_COMPRESSED = 11
MSG_RAW_PCAP = 12
MSG_PROFILE = 13
MSG_PROC_EVENT = 14
MSG_ALERT_RECORD = 15
MSG_K8S_EVENT = 16
MSG_APPLICATION_LOG = 17
MSG_AGENT_LOG = 18
MSG_SKYWALKING = 19
MSG_DATADOG = 20
MSG_ALERT_EVENT = 21

ENCODER_RAW = 0
ENCODER_ZSTD = 3


@dataclass
class FrameHeader:
    msg_type: int
    team_id: int = 0
    org_id: int = 1
    agent_id: int = 1
    encoder: int = ENCODER_RAW
    version: int = VERSION


def encode_frame(header: FrameHeader, payload: bytes) -> bytes:
    size = HEADER_LEN + len(payload)
    return b"".join((
        struct.pack(">I", size),
        struct.pack("<BHB", header.msg_type, header.version, header.encoder),
        struct.pack("<IHHHB", header.team_id, header.org_id, 0, header.agent_id, 0),
        payload,
    ))


def decode_frame(buf: bytes) -> Tuple[FrameHeader, bytes, int]:
    """Decode one frame; returns (header, payload, total_bytes_consumed)."""
    if len(buf) < HEADER_LEN:
        raise ValueError("short frame header")
    (size,) = struct.unpack_from(">I", buf, 0)
    if len(buf) < size:
        raise ValueError(f"short frame: want {size}, have {len(buf)}")
    msg_type, version, encoder = struct.unpack_from("<BHB", buf, 4)
    team_id, org_id, _, agent_id, _ = struct.unpack_from("<IHHHB", buf, 8)
    hdr = FrameHeader(msg_type=msg_type, team_id=team_id, org_id=org_id,
                      agent_id=agent_id, encoder=encoder, version=version)
    return hdr, bytes(buf[HEADER_LEN:size]), size


def pack_records(records: Iterable[bytes]) -> bytes:
    """Length-prefix each pb record (SimpleEncoder WritePB framing)."""
    parts: List[bytes] = []
    for r in records:
        parts.append(struct.pack("<I", len(r)))
        parts.append(r)
    return b"".join(parts)


def iter_records(payload: bytes) -> Iterator[bytes]:
    pos, end = 0, len(payload)
    mv = memoryview(payload)
    while pos + 4 <= end:
        (ln,) = struct.unpack_from("<I", payload, pos)
        pos += 4
        if pos + ln > end:
            raise ValueError("truncated record")
        yield bytes(mv[pos:pos + ln])
        pos += ln


def scan_record_offsets(payload: bytes) -> List[Tuple[int, int]]:
    """Return [(offset, len)] of each pb record inside a raw payload.

    The GPU ingest path uses the C++ twin of this (dfcpu_scan_offsets) to
    pre-segment records before handing the batch to the decode kernel.
    """
    out: List[Tuple[int, int]] = []
    pos, end = 0, len(payload)
    while pos + 4 <= end:
        (ln,) = struct.unpack_from("<I", payload, pos)
        pos += 4
        if pos + ln > end:
            raise ValueError("truncated record")
        out.append((pos, ln))
        pos += ln
    return out
