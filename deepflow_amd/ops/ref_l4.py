"""CPU reference for the L4 (TaggedFlow) decode + network.1s rollup.

Uses the generic pb decoder as an independent oracle (unlike ref.py's
hand-rolled L7 parser) — the GPU kernel is validated against a completely
separate implementation path.
"""
from __future__ import annotations

from typing import Dict, List

from ..store import l4_schema as L4
from ..store import l7_schema as S
from ..wire import pb, flow_log
from ..wire.pb import read_varint

M64 = (1 << 64) - 1

_U64 = {c: i for i, c in enumerate(L4.U64_COLS)}
_U32 = {c: i for i, c in enumerate(L4.U32_COLS)}
_U8 = {c: i for i, c in enumerate(L4.U8_COLS)}


def _find_str_refs(mv: memoryview, pos: int, end: int) -> Dict[int, tuple]:
    """Byte ranges of the pooled L4 strings: Flow.request_domain
    (path 1 -> 26) and FlowKey.ip6_src/dst (path 1 -> 1 -> 8/9).
    -> {str_col_idx: (off, len)}"""
    out: Dict[int, tuple] = {}
    while pos < end:
        key, pos = read_varint(mv, pos)
        num, wt = key >> 3, key & 7
        if num == 1 and wt == 2:
            ln, pos = read_varint(mv, pos)
            fend = pos + ln
            while pos < fend:
                k2, pos = read_varint(mv, pos)
                n2, w2 = k2 >> 3, k2 & 7
                if w2 == 2:
                    l2, pos = read_varint(mv, pos)
                    if n2 == 26:
                        out[L4.STR_COLS.index("request_domain")] = (pos, l2)
                    elif n2 == 1:  # FlowKey
                        p3, e3 = pos, pos + l2
                        while p3 < e3:
                            k3, p3 = read_varint(mv, p3)
                            n3, w3 = k3 >> 3, k3 & 7
                            if w3 == 2:
                                l3, p3 = read_varint(mv, p3)
                                if n3 == 8:
                                    out[L4.STR_COLS.index("ip6_0")] = (p3, l3)
                                elif n3 == 9:
                                    out[L4.STR_COLS.index("ip6_1")] = (p3, l3)
                                p3 += l3
                            elif w3 == 0:
                                _, p3 = read_varint(mv, p3)
                            elif w3 == 1:
                                p3 += 8
                            else:
                                p3 += 4
                    pos += l2
                elif w2 == 0:
                    _, pos = read_varint(mv, pos)
                elif w2 == 1:
                    pos += 8
                else:
                    pos += 4
            return out
        if wt == 0:
            _, pos = read_varint(mv, pos)
        elif wt == 2:
            ln, pos = read_varint(mv, pos)
            pos += ln
        elif wt == 1:
            pos += 8
        else:
            pos += 4
    return out


def decode_l4_ref(payload: bytes, offs, lens, seg, base_row: int,
                  sstr=None) -> None:
    mv = memoryview(payload)
    if sstr is None:
        import torch
        sstr = torch.zeros((L4.N_STR, len(offs)), dtype=torch.int64)
    for rid in range(len(offs)):
        row = base_row + rid
        off, ln = int(offs[rid]), int(lens[rid])
        d = pb.decode(mv[off:off + ln], flow_log.TAGGED_FLOW)
        f = d.get("flow", {})
        fk = f.get("flow_key", {})
        src = f.get("metrics_peer_src", {})
        dst = f.get("metrics_peer_dst", {})
        perf = f.get("perf_stats", {})
        tcp = perf.get("tcp", {})
        l7 = perf.get("l7", {})
        ptx = tcp.get("counts_peer_tx", {})
        prx = tcp.get("counts_peer_rx", {})

        def w64(col, v):
            seg.u64[_U64[col], row] = (v & M64) - (1 << 64) \
                if (v & M64) >= (1 << 63) else v & M64

        def w32(col, v):
            v &= 0xFFFFFFFF
            seg.u32[_U32[col], row] = v - (1 << 32) if v >= (1 << 31) else v

        def w8(col, v):
            seg.u8[_U8[col], row] = v & 0xFF

        w64("start_time", f.get("start_time", 0))
        w64("end_time", f.get("end_time", 0))
        w64("duration", f.get("duration", 0))
        w64("flow_id", f.get("flow_id", 0))
        w64("mac_src", fk.get("mac_src", 0))
        w64("mac_dst", fk.get("mac_dst", 0))
        for name, m in (("tx", src), ("rx", dst)):
            w64(f"byte_{name}", m.get("byte_count", 0))
            w64(f"l3_byte_{name}", m.get("l3_byte_count", 0))
            w64(f"l4_byte_{name}", m.get("l4_byte_count", 0))
            w64(f"packet_{name}", m.get("packet_count", 0))
            w64(f"total_byte_{name}", m.get("total_byte_count", 0))
            w64(f"total_packet_{name}", m.get("total_packet_count", 0))
        w64("l7_rrt_sum", l7.get("rrt_sum", 0))
        w32("vtap_id", fk.get("vtap_id", 0))
        w32("ip4_0", fk.get("ip_src", 0))
        w32("ip4_1", fk.get("ip_dst", 0))
        w32("l3_epc_id_0", src.get("l3_epc_id", 0))
        w32("l3_epc_id_1", dst.get("l3_epc_id", 0))
        w32("client_port", fk.get("port_src", 0))
        w32("server_port", fk.get("port_dst", 0))
        w32("tcp_flags_bit_0", src.get("tcp_flags", 0))
        w32("tcp_flags_bit_1", dst.get("tcp_flags", 0))
        w32("rtt", tcp.get("rtt", 0))
        for p in ("srt", "art", "cit"):
            w32(f"{p}_sum", tcp.get(f"{p}_sum", 0))
            w32(f"{p}_count", tcp.get(f"{p}_count", 0))
            w32(f"{p}_max", tcp.get(f"{p}_max", 0))
        w32("retrans_tx", ptx.get("retrans_count", 0))
        w32("retrans_rx", prx.get("retrans_count", 0))
        w32("zero_win_tx", ptx.get("zero_win_count", 0))
        w32("zero_win_rx", prx.get("zero_win_count", 0))
        w32("ooo_tx", ptx.get("ooo_count", 0))
        w32("ooo_rx", prx.get("ooo_count", 0))
        w32("syn_count", tcp.get("syn_count", 0))
        w32("synack_count", tcp.get("synack_count", 0))
        w32("retrans_total", tcp.get("total_retrans_count", 0))
        w32("l7_request", l7.get("request_count", 0))
        w32("l7_response", l7.get("response_count", 0))
        w32("l7_rrt_count", l7.get("rrt_count", 0))
        w32("l7_rrt_max", l7.get("rrt_max", 0))
        w32("l7_err_client", l7.get("err_client_count", 0))
        w32("l7_err_server", l7.get("err_server_count", 0))
        w32("l7_err_timeout", l7.get("err_timeout", 0))
        w32("gprocess_id_0", src.get("gpid", 0))
        w32("gprocess_id_1", dst.get("gpid", 0))
        w32("nat_real_ip_0", src.get("real_ip", 0))
        w32("nat_real_ip_1", dst.get("real_ip", 0))
        w32("nat_real_port_0", src.get("real_port", 0))
        w32("nat_real_port_1", dst.get("real_port", 0))
        w32("vlan", f.get("vlan", 0))
        w32("eth_type", f.get("eth_type", 0))
        gids = f.get("acl_gids", [])
        w32("acl_gid", gids[0] if gids else 0)
        w8("close_type", f.get("close_type", 0))
        w8("tap_side", f.get("tap_side", 0))
        w8("tap_type", fk.get("tap_type", 0))
        w8("protocol", fk.get("proto", 0))
        w8("l4_protocol", perf.get("l4_protocol", 0))
        w8("l7_protocol", perf.get("l7_protocol", 0))
        w8("signal_source", f.get("signal_source", 0))
        w8("is_new_flow", f.get("is_new_flow", 0))
        w8("is_active_service", f.get("is_active_service", 0))
        w8("direction_score", f.get("direction_score", 0))
        for sc, (roff, rlen) in _find_str_refs(mv, off, off + ln).items():
            sstr[sc, rid] = S.str_ref_pack(roff, rlen)
