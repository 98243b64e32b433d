"""CPU reference implementations of the HIP kernels.

Used (a) by tests on GPU-less machines, (b) as the numerics oracle the GPU
kernels are compared against (tests/test_gpu_pipeline.py), (c) by the
device='cpu' pipeline (multi-process gloo tests). Semantics mirror
ops/csrc/dfgpu.hip exactly; dictionary/table slot ids may differ from a
concurrent GPU run under hash collisions, so cross-checks compare hydrated
strings, not raw slot ids.
"""
from __future__ import annotations

from typing import Dict, List, Tuple

import numpy as np
import torch

from ..store import l7_schema as S
from ..store.dictionary import str_hash_py, domain_seed
from ..wire.pb import read_varint

M64 = (1 << 64) - 1
_HEXSET = frozenset(b"0123456789abcdefABCDEF")


def mix64(z: int) -> int:
    z &= M64
    z = ((z ^ (z >> 30)) * 0xBF58476D1CE4E5B9) & M64
    z = ((z ^ (z >> 27)) * 0x94D049BB133111EB) & M64
    return (z ^ (z >> 31)) & M64


# ---------------------------------------------------------------- K1 decode

_U64_IDX = {c: i for i, c in enumerate(S.U64_COLS)}
_U32_IDX = {c: i for i, c in enumerate(S.U32_COLS)}
_U8_IDX = {c: i for i, c in enumerate(S.U8_COLS)}
_STR_IDX = {c: i for i, c in enumerate(S.STR_COLS)}

_BASE_U = {
    1: ("u64", "start_time"), 2: ("u64", "end_time"), 3: ("u64", "flow_id"),
    5: ("u32", "vtap_id"), 6: ("u8", "tap_type"), 7: ("u8", "is_ipv6"),
    8: ("u8", "tap_side"), 12: ("u32", "ip4_0"), 13: ("u32", "ip4_1"),
    16: ("u32", "l3_epc_id_0"), 17: ("u32", "l3_epc_id_1"),
    18: ("u32", "client_port"), 19: ("u32", "server_port"),
    20: ("u8", "protocol"), 23: ("u32", "req_tcp_seq"),
    24: ("u32", "resp_tcp_seq"), 25: ("u32", "process_id_0"),
    26: ("u32", "process_id_1"), 29: ("u64", "syscall_trace_id_request"),
    30: ("u64", "syscall_trace_id_response"), 35: ("u32", "gprocess_id_0"),
    36: ("u32", "gprocess_id_1"), 41: ("u32", "pod_id_0"),
    42: ("u32", "pod_id_1"), 43: ("u32", "biz_type"),
}


def _w(seg, fam, col, row, v):
    t = {"u64": seg.u64, "u32": seg.u32, "u8": seg.u8}[fam]
    idx = {"u64": _U64_IDX, "u32": _U32_IDX, "u8": _U8_IDX}[fam][col]
    if fam == "u64":
        t[idx, row] = v & M64 if v < (1 << 63) else (v & M64) - (1 << 64)
    elif fam == "u32":
        v32 = v & 0xFFFFFFFF
        t[idx, row] = v32 if v32 < (1 << 31) else v32 - (1 << 32)
    else:
        t[idx, row] = v & 0xFF


def decode_l7_ref(payload: bytes, offs, lens, seg, base_row: int,
                  sstr=None, sattr=None) -> None:
    mv = memoryview(payload)
    if sstr is None:
        import torch
        sstr = torch.zeros((S.N_STR, len(offs)), dtype=torch.int64)
        sattr = torch.zeros((2 * S.MAX_ATTRS, len(offs)), dtype=torch.int64)
    seg._last_sstr, seg._last_sattr = sstr, sattr
    for rid in range(len(offs)):
        row = base_row + rid
        pos = int(offs[rid])
        end = pos + int(lens[rid])
        n_names = n_vals = 0
        while pos < end:
            key, pos = read_varint(mv, pos)
            num, wt = key >> 3, key & 7
            if wt == 0:
                v, pos = read_varint(mv, pos)
                if num == 9:
                    _w(seg, "u32", "request_length", row, v)
                elif num == 10:
                    _w(seg, "u32", "response_length", row, v)
                elif num == 17:
                    _w(seg, "u8", "direction_score", row, v)
                elif num == 18:
                    _w(seg, "u32", "flags", row, v)
                elif num == 19:
                    _w(seg, "u32", "captured_request_byte", row, v)
                elif num == 20:
                    _w(seg, "u32", "captured_response_byte", row, v)
            elif wt == 2:
                ln, pos = read_varint(mv, pos)
                sub, send = pos, pos + ln
                pos = send
                if num == 1:  # base
                    p2 = sub
                    while p2 < send:
                        k2, p2 = read_varint(mv, p2)
                        n2, w2 = k2 >> 3, k2 & 7
                        if w2 == 0:
                            v, p2 = read_varint(mv, p2)
                            if n2 in _BASE_U:
                                fam, col = _BASE_U[n2]
                                _w(seg, fam, col, row, v)
                        elif w2 == 2:
                            l3, p2 = read_varint(mv, p2)
                            s3, e3 = p2, p2 + l3
                            p2 = e3
                            if n2 == 9:  # head
                                p3 = s3
                                while p3 < e3:
                                    k3, p3 = read_varint(mv, p3)
                                    if (k3 & 7) == 0:
                                        v, p3 = read_varint(mv, p3)
                                        if k3 >> 3 == 1:
                                            _w(seg, "u8", "l7_protocol", row, v)
                                        elif k3 >> 3 == 2:
                                            _w(seg, "u8", "msg_type", row, v)
                                        elif k3 >> 3 == 5:
                                            _w(seg, "u64", "rrt", row, v)
                                    else:
                                        p3 = e3
                            elif n2 == 27:
                                sstr[_STR_IDX["process_kname_0"], rid] = \
                                    S.str_ref_pack(s3, l3)
                            elif n2 == 28:
                                sstr[_STR_IDX["process_kname_1"], rid] = \
                                    S.str_ref_pack(s3, l3)
                            elif n2 == 14:  # ip6_src (raw 16 bytes)
                                sstr[_STR_IDX["ip6_0"], rid] = \
                                    S.str_ref_pack(s3, l3)
                            elif n2 == 15:  # ip6_dst
                                sstr[_STR_IDX["ip6_1"], rid] = \
                                    S.str_ref_pack(s3, l3)
                        elif w2 == 1:
                            p2 += 8
                        elif w2 == 5:
                            p2 += 4
                elif num in (11, 12, 14, 15):
                    strmap = {
                        11: {1: "request_type", 2: "request_domain",
                             3: "request_resource", 4: "endpoint"},
                        12: {3: "exception_desc", 4: "response_result"},
                        14: {1: "trace_id", 2: "span_id", 3: "parent_span_id"},
                        15: {1: "service_name", 4: "x_request_id_0",
                             6: "http_user_agent", 7: "http_referer",
                             10: "x_request_id_1"},
                    }[num]
                    p2 = sub
                    while p2 < send:
                        k2, p2 = read_varint(mv, p2)
                        n2, w2 = k2 >> 3, k2 & 7
                        if w2 == 0:
                            v, p2 = read_varint(mv, p2)
                            if num == 12 and n2 == 1:
                                _w(seg, "u8", "response_status", row, v)
                            elif num == 12 and n2 == 2:
                                _w(seg, "u32", "response_code", row, v)
                            elif num == 15 and n2 == 3:
                                _w(seg, "u32", "request_id", row, v)
                        elif w2 == 2:
                            l3, p2 = read_varint(mv, p2)
                            if num == 15 and n2 == 16:
                                if n_names < S.MAX_ATTRS:
                                    sattr[n_names, rid] = \
                                        S.str_ref_pack(p2, l3)
                                n_names += 1
                            elif num == 15 and n2 == 17:
                                if n_vals < S.MAX_ATTRS:
                                    sattr[S.MAX_ATTRS + n_vals, rid] = \
                                        S.str_ref_pack(p2, l3)
                                n_vals += 1
                            elif num == 14 and n2 in (1, 2):
                                # hex trace/span ids -> binary u64 cols
                                # (pool fallback for non-hex forms)
                                raw = bytes(mv[p2:p2 + l3])
                                hexok = all(c in _HEXSET for c in raw)
                                done = False
                                if hexok and n2 == 1 and l3 == 32:
                                    v = int(raw, 16)
                                    _w(seg, "u64", "trace_id_hi", row,
                                       v >> 64)
                                    _w(seg, "u64", "trace_id_lo", row,
                                       v & M64)
                                    done = True
                                elif hexok and n2 == 2 and l3 == 16:
                                    _w(seg, "u64", "span_id_b", row,
                                       int(raw, 16))
                                    done = True
                                if not done:
                                    sstr[_STR_IDX[strmap[n2]], rid] = \
                                        S.str_ref_pack(p2, l3)
                            elif n2 in strmap:
                                sstr[_STR_IDX[strmap[n2]], rid] = \
                                    S.str_ref_pack(p2, l3)
                            p2 += l3
                        elif w2 == 1:
                            p2 += 8
                        elif w2 == 5:
                            p2 += 4
                elif num == 13:
                    sstr[_STR_IDX["version"], rid] = S.str_ref_pack(sub, ln)
                elif num == 21:
                    sstr[_STR_IDX["biz_code"], rid] = S.str_ref_pack(sub, ln)
            elif wt == 1:
                pos += 8
            elif wt == 5:
                pos += 4
        na = min(n_names, n_vals, S.MAX_ATTRS)
        seg.attr_cnt[row] = na


# ------------------------------------------------------------ K2 kg tables

def kg_build_ref(keys: torch.Tensor, vals: torch.Tensor,
                 tkeys: torch.Tensor, tvals: torch.Tensor) -> None:
    cap_mask = tkeys.numel() - 1
    tk = tkeys.numpy()
    tv = tvals.numpy()
    kk = keys.numpy().view(np.uint64)
    vv = vals.numpy()
    for i in range(len(kk)):
        k = int(kk[i])
        if k == 0:
            continue
        slot = mix64(k) & cap_mask
        for _ in range(cap_mask + 1):
            cur = int(tk[slot]) & M64
            if cur == 0 or cur == k:
                tk[slot] = np.int64(np.uint64(k).astype(np.int64)) \
                    if k < (1 << 63) else np.int64(k - (1 << 64))
                tv[slot] = vv[i]
                break
            slot = (slot + 1) & cap_mask



def intern_ref(payload: bytes, refs: torch.Tensor, ref_rows, domains,
               ref_base_row: int, n: int, tkeys: torch.Tensor,
               out_ids: torch.Tensor, out_base_row: int,
               dictionary=None) -> List[Tuple[int, int, bytes]]:
    """Sequential-deterministic intern; returns new (domain, slot, bytes)."""
    cap_mask = tkeys.numel() - 1
    tk = tkeys.numpy()
    new_entries: List[Tuple[int, int, bytes]] = []
    for ci, (rrow, dom) in enumerate(zip(ref_rows, domains)):
        for i in range(n):
            ref = int(refs[rrow, ref_base_row + i].item()) & M64
            ln = ref & 0xFFFF
            off = ref >> 16
            if ln == 0:
                out_ids[ci, out_base_row + i] = -1  # DICT_ID_INVALID as i32
                continue
            sbytes = payload[off:off + ln]
            h = str_hash_py(bytes(sbytes), domain_seed(dom))
            slot = h & cap_mask
            placed = False
            for _ in range(cap_mask + 1):
                cur = int(tk[slot]) & M64
                if cur == 0:
                    tk[slot] = np.int64(h) if h < (1 << 63) \
                        else np.int64(h - (1 << 64))
                    new_entries.append((dom, slot, bytes(sbytes)))
                    placed = True
                    break
                if cur == h:
                    placed = True
                    break
                slot = (slot + 1) & cap_mask
            v = slot if placed else S.DICT_ID_INVALID
            out_ids[ci, out_base_row + i] = v if v < (1 << 31) else v - (1 << 32)
    if dictionary is not None:
        for dom, slot, sbytes in new_entries:
            dictionary.id_to_str[(dom, slot)] = sbytes
            dictionary.str_to_id[(dom, sbytes)] = slot
            dictionary.pending_sync.append((dom, slot, sbytes))
    return new_entries


def intern_attrs_pool_ref(payload: bytes, sattr, seg, base_row: int, n: int,
                          tkeys, dictionary=None):
    """CPU twin of k_intern_attrs with the variable attr-id pool."""
    new_entries = []
    for i in range(n):
        row = base_row + i
        cnt = int(seg.attr_cnt[row].item())
        start = seg.attr_pool_len
        seg.ensure_attr_pool(2 * cnt)
        seg.attr_start[row] = start
        for a in range(cnt):
            for half, dom in ((0, S.DICT_DOM_ATTR_NAME),
                              (1, S.DICT_DOM_ATTR_VALUE)):
                ref = int(sattr[half * S.MAX_ATTRS + a, i].item()) & M64
                ln = ref & 0xFFFF
                off = ref >> 16
                ident = -1
                if ln:
                    sbytes = payload[off:off + ln]
                    h = str_hash_py(bytes(sbytes), domain_seed(dom))
                    cap_mask = tkeys.numel() - 1
                    tk = tkeys.numpy()
                    slot = h & cap_mask
                    for _ in range(cap_mask + 1):
                        cur = int(tk[slot]) & M64
                        if cur == 0:
                            tk[slot] = np.int64(h) if h < (1 << 63) \
                                else np.int64(h - (1 << 64))
                            new_entries.append((dom, slot, bytes(sbytes)))
                            break
                        if cur == h:
                            break
                        slot = (slot + 1) & cap_mask
                    ident = slot if slot < (1 << 31) else slot - (1 << 32)
                seg.attr_pool[start + half * cnt + a] = ident
        seg.attr_pool_len += 2 * cnt
    if dictionary is not None:
        for dom, slot, sbytes in new_entries:
            dictionary.id_to_str[(dom, slot)] = sbytes
            dictionary.str_to_id[(dom, sbytes)] = slot
            dictionary.pending_sync.append((dom, slot, sbytes))
    return new_entries


# -------------------------------------------------------------- K4 pool

def pool_lens_ref(sstr, pool_cols, n: int) -> torch.Tensor:
    out = torch.zeros(n, dtype=torch.int32)
    for i in range(n):
        total = 0
        for c in pool_cols:
            total += int(sstr[c, i].item()) & 0xFFFF
        out[i] = total
    return out


def pool_gather_ref(payload: bytes, seg, pool_cols, base_row: int, n: int,
                    row_start: torch.Tensor, pool_base: int, sstr) -> None:
    pool = seg.pool.numpy()
    for i in range(n):
        dst0 = pool_base + int(row_start[i].item())
        dst = dst0
        for ci, c in enumerate(pool_cols):
            r = int(sstr[c, i].item()) & M64
            ln = r & 0xFFFF
            off = r >> 16
            pool[dst:dst + ln] = np.frombuffer(payload[off:off + ln],
                                               dtype=np.uint8)
            seg.str_lens[ci, base_row + i] = ln
            dst += ln
        seg.str_rowref[base_row + i] = S.str_ref_pack(dst0, dst - dst0)


# -------------------------------------------------------------- K5 agg

AGG_NVALS = 7  # req, resp, err_c, err_s, rrt_sum, rrt_cnt, rrt_max


