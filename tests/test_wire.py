"""Wire layer tests: varint/pb codec, trident framing, generator roundtrips."""
import struct

import pytest

from deepflow_amd.wire import pb, flow_log, metric, framing
from deepflow_amd.gen import (
    SpanGenConfig, gen_span_dict, FlowGenConfig, gen_flow_dict,
    DocGenConfig, gen_document_dict,
)
from deepflow_amd.gen.spans import gen_span_records
from deepflow_amd.gen.flows import gen_flow_records
from deepflow_amd.gen.documents import gen_document_records


def test_varint_roundtrip():
    for v in [0, 1, 127, 128, 300, 2 ** 32 - 1, 2 ** 63, 2 ** 64 - 1]:
        buf = bytearray()
        pb.write_varint(buf, v)
        got, pos = pb.read_varint(memoryview(bytes(buf)), 0)
        assert got == v and pos == len(buf)


def test_varint_golden():
    # protobuf spec examples
    buf = bytearray()
    pb.write_varint(buf, 300)
    assert bytes(buf) == b"\xac\x02"


def test_negative_int32_ten_bytes():
    # proto3 int32 -2 encodes as 64-bit two's complement varint (10 bytes)
    schema = {1: ("x", 'i')}
    enc = pb.encode({"x": -2}, schema)
    assert len(enc) == 1 + 10
    assert pb.decode(enc, schema) == {"x": -2}


def test_packed_repeated():
    schema = {3: ("v", '*u')}
    enc = pb.encode({"v": [3, 270, 86942]}, schema)
    # field 3, wiretype 2 (packed)
    assert enc[0] == (3 << 3) | 2
    assert pb.decode(enc, schema) == {"v": [3, 270, 86942]}


def test_repeated_string():
    schema = {2: ("names", '*s')}
    enc = pb.encode({"names": ["a", "bc"]}, schema)
    assert pb.decode(enc, schema) == {"names": ["a", "bc"]}


def test_span_roundtrip():
    cfg = SpanGenConfig(n=5, tag_cardinality=100)
    for i in range(5):
        d = gen_span_dict(cfg, i)
        enc = pb.encode(d, flow_log.APP_PROTO_LOGS_DATA)
        dec = pb.decode(enc, flow_log.APP_PROTO_LOGS_DATA)
        assert dec["base"]["start_time"] == d["base"]["start_time"]
        assert dec["req"]["domain"] == d["req"]["domain"]
        assert dec["trace_info"]["trace_id"] == d["trace_info"]["trace_id"]
        assert dec["ext_info"]["attribute_values"] == d["ext_info"]["attribute_values"]
        # proto3 zero-default fields are dropped on the wire
        if d["resp"]["status"] == 0:
            assert "status" not in dec["resp"]


def test_flow_roundtrip():
    cfg = FlowGenConfig(n=3)
    for i in range(3):
        d = gen_flow_dict(cfg, i)
        enc = pb.encode(d, flow_log.TAGGED_FLOW)
        dec = pb.decode(enc, flow_log.TAGGED_FLOW)
        assert dec["flow"]["flow_id"] == d["flow"]["flow_id"]
        assert dec["flow"]["metrics_peer_src"]["byte_count"] == \
            d["flow"]["metrics_peer_src"]["byte_count"]
        assert dec["flow"]["perf_stats"]["tcp"]["rtt"] == \
            d["flow"]["perf_stats"]["tcp"]["rtt"]


def test_document_roundtrip():
    cfg = DocGenConfig(n=3)
    for i in range(3):
        d = gen_document_dict(cfg, i)
        enc = pb.encode(d, metric.DOCUMENT)
        dec = pb.decode(enc, metric.DOCUMENT)
        assert dec["timestamp"] == d["timestamp"]
        assert dec["meter"]["app"]["traffic"]["request"] == \
            d["meter"]["app"]["traffic"]["request"]
        assert dec["tag"]["field"]["ip"] == d["tag"]["field"]["ip"]


def test_frame_roundtrip():
    hdr = framing.FrameHeader(msg_type=framing.MSG_PROTOCOLLOG, team_id=7,
                              org_id=2, agent_id=42)
    payload = framing.pack_records([b"abc", b"", b"defg"])
    frame = framing.encode_frame(hdr, payload)
    # golden header layout
    assert frame[0:4] == struct.pack(">I", len(frame))
    assert frame[4] == framing.MSG_PROTOCOLLOG
    assert struct.unpack_from("<H", frame, 5)[0] == 0x8000
    assert frame[7] == framing.ENCODER_RAW
    assert struct.unpack_from("<I", frame, 8)[0] == 7
    assert struct.unpack_from("<H", frame, 12)[0] == 2
    assert struct.unpack_from("<H", frame, 16)[0] == 42
    h2, p2, consumed = framing.decode_frame(frame)
    assert consumed == len(frame)
    assert (h2.msg_type, h2.team_id, h2.org_id, h2.agent_id) == (5, 7, 2, 42)
    assert list(framing.iter_records(p2)) == [b"abc", b"", b"defg"]
    offs = framing.scan_record_offsets(p2)
    assert [p2[o:o + l] for o, l in offs] == [b"abc", b"", b"defg"]


def test_generators_deterministic():
    a = gen_span_records(SpanGenConfig(n=4))
    b = gen_span_records(SpanGenConfig(n=4))
    assert a == b
    assert gen_flow_records(FlowGenConfig(n=4)) == gen_flow_records(FlowGenConfig(n=4))
    assert gen_document_records(DocGenConfig(n=4)) == \
        gen_document_records(DocGenConfig(n=4))
