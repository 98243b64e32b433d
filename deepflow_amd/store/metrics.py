"""flow_metrics table family: exact-key GPU rollup tables.

Reference counterpart: server/libs/flow-metrics (tag.go:443-523 — the 9
tables network{,_map}.{1s,1m}, application{,_map}.{1s,1m},
traffic_policy.1m, each selecting its tag columns with a Code bitmask)
plus ingester/flow_metrics/unmarshaller routing documents into them.

MI355X design: each table is an open-addressing device-resident hash
aggregation — tkeys[cap] holds the 64-bit mixed hash of the key tuple
(the CAS claim word), traw[cap][8] the raw tuple for verification and
harvest, tvals[cap][nv] the accumulators (dfgpu.hip k_rollup_l4 /
k_rollup_l7 / k_rollup_insert). Keys are EXACT: no masked bit-packing, so
distinct groups never merge (round-1 defect). Harvest merges the rare
duplicate slots created by the race-free claim protocol.

CPU mode mirrors the semantics with a plain dict keyed by the raw tuple —
the numerics oracle for the GPU tests.
"""
from __future__ import annotations

import struct

import numpy as np
from dataclasses import dataclass
from typing import Dict, List, Optional, Tuple

import torch

RU_MAX_KEYS = 8  # mirrors dfgpu.hip

FAM_U64, FAM_U32, FAM_U8 = 0, 1, 2

NET_FIELDS = ["byte_tx", "byte_rx", "packet_tx", "packet_rx", "new_flow",
              "closed_flow", "rtt_sum", "rtt_count", "rtt_max", "retrans"]
APP_FIELDS = ["request", "response", "client_error", "server_error",
              "rrt_sum", "rrt_count", "rrt_max"]
# per-field merge op for harvest / insert (0=sum, 1=max)
NET_OPS = [0] * 8 + [1, 0]
APP_OPS = [0] * 6 + [1]

# Code bitmask (reference libs/flow-metrics/tag.go Code composition);
# documents which tag columns each table's key carries
CODE_IP = 1 << 0
CODE_L3_EPC = 1 << 1
CODE_VTAP = 1 << 2
CODE_PROTOCOL = 1 << 3
CODE_SERVER_PORT = 1 << 4
CODE_TAP_TYPE = 1 << 5
CODE_L7_PROTOCOL = 1 << 6
CODE_STATUS = 1 << 7
CODE_IP_PATH = 1 << 8     # _map tables: both endpoints
CODE_ACL_GID = 1 << 9


@dataclass(frozen=True)
class TableDef:
    name: str
    source: str               # 'l4' | 'l7'
    interval_s: int
    keys: Tuple[str, ...]     # segment column names (order = key order)
    agg: str                  # 'net' | 'app'
    code: int
    require_nonzero: Optional[str] = None  # key that must be != 0
    # output name per key (reference tag naming); None = same as key col
    out_names: Optional[Tuple[str, ...]] = None
    # device table capacity hint (log2); _map tables carry endpoint pairs
    # and need far more slots than the scalar-tag tables
    cap_pow2: int = 18
    # harvest-time fold: this table's rows derive from another table's
    # harvest (1m = fold(1s, 60)) instead of per-row kernel updates —
    # the per-row path costs millions of random atomics, the fold costs
    # thousands of group merges
    derive_from: Optional[str] = None


def _net(name, interval, map_):
    if map_:
        return TableDef(
            name, "l4", interval,
            ("vtap_id", "ip4_0", "ip4_1", "l3_epc_id_0", "l3_epc_id_1",
             "protocol", "server_port"), "net",
            CODE_VTAP | CODE_IP_PATH | CODE_L3_EPC | CODE_PROTOCOL |
            CODE_SERVER_PORT,
            out_names=("vtap_id", "ip_0", "ip_1", "l3_epc_id_0",
                       "l3_epc_id_1", "protocol", "server_port"),
            cap_pow2=21)
    return TableDef(
        name, "l4", interval,
        ("vtap_id", "l3_epc_id_0", "protocol", "server_port", "tap_type"),
        "net",
        CODE_VTAP | CODE_L3_EPC | CODE_PROTOCOL | CODE_SERVER_PORT |
        CODE_TAP_TYPE,
        out_names=("vtap_id", "l3_epc_id", "protocol", "server_port",
                   "tap_type"))


def _app(name, interval, map_):
    if map_:
        return TableDef(
            name, "l7", interval,
            ("vtap_id", "ip4_0", "ip4_1", "l7_protocol", "server_port"),
            "app",
            CODE_VTAP | CODE_IP_PATH | CODE_L7_PROTOCOL | CODE_SERVER_PORT,
            out_names=("vtap_id", "ip_0", "ip_1", "l7_protocol",
                       "server_port"), cap_pow2=21)
    return TableDef(
        name, "l7", interval,
        ("vtap_id", "l7_protocol", "response_status", "server_port"), "app",
        CODE_VTAP | CODE_L7_PROTOCOL | CODE_STATUS | CODE_SERVER_PORT)


import dataclasses as _dc

L4_TABLES = [
    _net("network.1s", 1, False),
    _dc.replace(_net("network.1m", 60, False),
                derive_from="network.1s"),
    _net("network_map.1s", 1, True),
    _dc.replace(_net("network_map.1m", 60, True),
                derive_from="network_map.1s"),
    # traffic_policy rows exist only for ACL-matched flows (tiny
    # fraction) — the per-row path is already cheap there
    TableDef("traffic_policy.1m", "l4", 60, ("vtap_id", "acl_gid"), "net",
             CODE_VTAP | CODE_ACL_GID, require_nonzero="acl_gid"),
]
L7_TABLES = [
    _app("application.1s", 1, False),
    _dc.replace(_app("application.1m", 60, False),
                derive_from="application.1s"),
    _app("application_map.1s", 1, True),
    _dc.replace(_app("application_map.1m", 60, True),
                derive_from="application_map.1s"),
]


def _resolve(source: str, col: str) -> Tuple[int, int]:
    if source == "l4":
        from . import l4_schema as M
    else:
        from . import l7_schema as M
    if col in M.U64_COLS:
        return FAM_U64, M.U64_COLS.index(col)
    if col in M.U32_COLS:
        return FAM_U32, M.U32_COLS.index(col)
    if col in M.U8_COLS:
        return FAM_U8, M.U8_COLS.index(col)
    raise KeyError(f"{source} column {col!r} not in fixed-width schema")


M32 = 0xFFFFFFFF
M64 = (1 << 64) - 1


def _np_merge(keys: "np.ndarray", vals: "np.ndarray", ops):
    """Merge duplicate key rows: sum columns with op 0, max otherwise.
    lexsort + reduceat over the sorted runs — np.unique(axis=0) +
    ufunc.at measured ~20x slower on multi-million-group tables."""
    if keys.shape[0] == 0:
        return keys, vals
    order = np.lexsort(keys.T[::-1])
    ks = keys[order]
    vs = vals[order]
    change = np.any(ks[1:] != ks[:-1], axis=1)
    starts = np.concatenate(([0], np.nonzero(change)[0] + 1))
    uniq = ks[starts]
    out = np.empty((len(starts), vs.shape[1]), dtype=np.uint64)
    for v in range(vs.shape[1]):
        col = vs[:, v]
        if ops[v] == 0:
            out[:, v] = np.add.reduceat(col, starts)
        else:
            out[:, v] = np.maximum.reduceat(col, starts)
    return uniq, out


class RollupTable:
    def __init__(self, td: TableDef, time_base_s: int, device: str = "cpu",
                 capacity_pow2: int = 1 << 18):
        assert capacity_pow2 & (capacity_pow2 - 1) == 0
        assert len(td.keys) <= RU_MAX_KEYS - 1
        self.td = td
        self.time_base_s = time_base_s
        self.device = device
        self.capacity = capacity_pow2
        self.fields = NET_FIELDS if td.agg == "net" else APP_FIELDS
        self.ops = NET_OPS if td.agg == "net" else APP_OPS
        self.nv = len(self.fields)
        self.nw = len(td.keys) + 1  # + time bucket word
        # source 'doc': insert-only table (pre-aggregated agent Documents);
        # no segment columns to resolve
        self._srcs = ([] if td.source == "doc"
                      else [_resolve(td.source, c) for c in td.keys])
        self._rnz = (1 + td.keys.index(td.require_nonzero)
                     if td.require_nonzero else 0)
        self.source_table = None  # set by RollupFamily for derive_from
        # interval flush (reference: each 1s/1m interval is written out
        # to ClickHouse and the in-memory aggregator moves on): the live
        # device/CPU table archives-and-resets periodically so bounded
        # caps never saturate on long runs. Archived groups are COMPACT
        # numpy (keys, vals) chunks — a _map flush can carry millions of
        # groups, far beyond what per-row python dicts can shuffle —
        # and merge with live groups at read time.
        self.archive: List[tuple] = []  # [(keys u64 [G,nw], vals [G,nv])]
        self.dropped_total = 0
        if td.derive_from is not None:
            return
        if device == "cpu":
            self.table: Dict[tuple, List[int]] = {}
        else:
            dev = torch.device(device)
            self.tkeys = torch.zeros(capacity_pow2, dtype=torch.int64,
                                     device=dev)
            self.traw = torch.full((capacity_pow2, RU_MAX_KEYS), -1,
                                   dtype=torch.int64, device=dev)
            self.tvals = torch.zeros((capacity_pow2, self.nv),
                                     dtype=torch.int64, device=dev)
            self.drops = torch.zeros(1, dtype=torch.int64, device=dev)

    # RuSpec ABI twin (dfgpu.hip): {u32 interval; u32 n_keys; u8 fam[8];
    # u8 idx[8]; u8 require_nonzero; u8 use_lds} padded to 4-byte alignment
    def spec_bytes(self) -> bytes:
        fam = [f for f, _ in self._srcs] + [0] * (RU_MAX_KEYS - len(self._srcs))
        idx = [i for _, i in self._srcs] + [0] * (RU_MAX_KEYS - len(self._srcs))
        # LDS pre-aggregation pays off on low-cardinality tables (hot
        # global slots); the _map tables (cap >= 2^21) mostly miss the
        # 256-slot block table and go direct
        use_lds = 1 if self.td.cap_pow2 < 21 else 0
        return struct.pack("<II8B8BBB2x", self.td.interval_s,
                           len(self._srcs), *fam, *idx, self._rnz, use_lds)

    # ----------------------------------------------------------- update
    def update(self, seg, base: int, n: int, stream: int = 0) -> None:
        if n == 0:
            return
        if self.device == "cpu":
            self._update_cpu(seg, base, n)
        else:
            from ..ops import gpu_ops
            gpu_ops.rollup_family(seg, base, n, self.time_base_s, [self],
                                  stream=stream)

    def _key_tuple(self, seg, row: int) -> Optional[tuple]:
        t_s = (int(seg.u64[0, row].item()) & M64) // 10**9
        rel = max(t_s - self.time_base_s, 0)
        iv = self.td.interval_s
        if iv > 1:
            rel = (rel // iv) * iv
        kw = [rel]
        for fam, idx in self._srcs:
            if fam == FAM_U64:
                kw.append(int(seg.u64[idx, row].item()) & M64)
            elif fam == FAM_U32:
                kw.append(int(seg.u32[idx, row].item()) & M32)
            else:
                kw.append(int(seg.u8[idx, row].item()) & 0xFF)
        if self._rnz and kw[self._rnz] == 0:
            return None
        return tuple(kw)

    def _update_cpu(self, seg, base: int, n: int) -> None:
        from . import l4_schema as L4
        from . import l7_schema as L7
        M = L4 if self.td.source == "l4" else L7
        u64i = {c: i for i, c in enumerate(M.U64_COLS)}
        u32i = {c: i for i, c in enumerate(M.U32_COLS)}
        u8i = {c: i for i, c in enumerate(M.U8_COLS)}
        for i in range(n):
            row = base + i
            key = self._key_tuple(seg, row)
            if key is None:
                continue
            acc = self.table.setdefault(key, [0] * self.nv)
            if self.td.agg == "net":
                acc[0] += int(seg.u64[u64i["byte_tx"], row].item()) & M64
                acc[1] += int(seg.u64[u64i["byte_rx"], row].item()) & M64
                acc[2] += int(seg.u64[u64i["packet_tx"], row].item()) & M64
                acc[3] += int(seg.u64[u64i["packet_rx"], row].item()) & M64
                if int(seg.u8[u8i["is_new_flow"], row].item()):
                    acc[4] += 1
                if int(seg.u8[u8i["close_type"], row].item()):
                    acc[5] += 1
                rtt = int(seg.u32[u32i["rtt"], row].item()) & M32
                if rtt:
                    acc[6] += rtt
                    acc[7] += 1
                    acc[8] = max(acc[8], rtt)
                retrans = (int(seg.u32[u32i["retrans_tx"], row].item()) & M32) + \
                    (int(seg.u32[u32i["retrans_rx"], row].item()) & M32)
                if retrans:
                    acc[9] += retrans
            else:
                mtype = int(seg.u8[u8i["msg_type"], row].item())
                status = int(seg.u8[u8i["response_status"], row].item())
                rrt = int(seg.u64[u64i["rrt"], row].item()) & M64
                if mtype in (0, 2):
                    acc[0] += 1
                if mtype in (1, 2):
                    acc[1] += 1
                if status == 4:
                    acc[2] += 1
                if status == 3:
                    acc[3] += 1
                if rrt:
                    acc[4] += rrt
                    acc[5] += 1
                    acc[6] = max(acc[6], rrt)

    # ------------------------------------------------------------ insert
    def insert(self, key_rows: List[tuple], val_rows: List[List[int]],
               stream: int = 0) -> None:
        """Merge pre-aggregated (key tuple, values) rows — the agent
        Document ingest path (reference flow_metrics unmarshaller)."""
        if not key_rows:
            return
        if self.device == "cpu":
            for kw, vals in zip(key_rows, val_rows):
                acc = self.table.setdefault(tuple(kw), [0] * self.nv)
                for v in range(self.nv):
                    if self.ops[v] == 0:
                        acc[v] += vals[v]
                    else:
                        acc[v] = max(acc[v], vals[v])
            return
        from ..ops import gpu_ops
        dev = torch.device(self.device)
        kws = torch.tensor([list(k) + [0] * (self.nw - len(k))
                            for k in key_rows],
                           dtype=torch.int64, device=dev)[:, : self.nw]
        vals = torch.tensor(val_rows, dtype=torch.int64, device=dev)
        ops = torch.tensor(self.ops, dtype=torch.uint8, device=dev)
        gpu_ops.rollup_insert(kws, vals, ops, self, stream=stream)

    # ------------------------------------------------------------ harvest
    def _harvest_np(self):
        """(keys u64 [G, nw], vals u64 [G, nv]) of the LIVE table with
        duplicate-slot groups merged — all numpy, no per-row python."""
        if self.device == "cpu":
            if not self.table:
                return (np.zeros((0, self.nw), dtype=np.uint64),
                        np.zeros((0, self.nv), dtype=np.uint64))
            keys = np.array(list(self.table.keys()), dtype=np.uint64)
            vals = np.array(list(self.table.values()), dtype=np.uint64)
            return keys.reshape(-1, self.nw), vals.reshape(-1, self.nv)
        mask = self.tkeys != 0
        raw = self.traw[mask][:, : self.nw].cpu().numpy().view(np.uint64)
        vals = self.tvals[mask].cpu().numpy().view(np.uint64)
        return raw, vals

    def _items(self) -> List[Tuple[tuple, List[int]]]:
        if self.device == "cpu":
            return [(k, list(v)) for k, v in self.table.items()]
        mask = self.tkeys != 0
        raw = self.traw[mask][:, : self.nw].cpu().numpy()
        vals = self.tvals[mask].cpu().numpy()
        merged: Dict[tuple, List[int]] = {}
        for r in range(raw.shape[0]):
            key = tuple(int(x) & M64 for x in raw[r])
            acc = merged.get(key)
            if acc is None:
                merged[key] = [int(v) & M64 for v in vals[r]]
            else:  # duplicate slot from the race-free claim: exact merge
                for v in range(self.nv):
                    if self.ops[v] == 0:
                        acc[v] += int(vals[r][v]) & M64
                    else:
                        acc[v] = max(acc[v], int(vals[r][v]) & M64)
        return list(merged.items())

    def _fmt_key(self, name: str, col: str, v: int):
        if name == "ip" or name.startswith("ip_"):
            return "%d.%d.%d.%d" % ((v >> 24) & 255, (v >> 16) & 255,
                                    (v >> 8) & 255, v & 255)
        if "epc" in name:        # epc ids are signed int32
            return v - (1 << 32) if v >= (1 << 31) else v
        return v

    def flush_live(self) -> int:
        """Archive the live table's groups (a raw compact dump — the
        hash table is already key-unique up to rare duplicate-claim
        slots, which the read-time merge folds) and reset it. Derived
        tables hold no state: they fold from the source's archive +
        live at read time, so flush skips them entirely — the flush
        cost is one D2H copy per table, not a host-side merge."""
        if self.td.derive_from is not None:
            return 0
        keys, vals = self._harvest_np()
        if keys.shape[0]:
            self.archive.append((keys, vals))
            if len(self.archive) > 8:
                # compact: one lexsort merge bounds archive memory and
                # read-time concat cost (groups recurring across flush
                # windows collapse)
                ak = np.concatenate([k for k, _ in self.archive], axis=0)
                av = np.concatenate([v for _, v in self.archive], axis=0)
                self.archive = [_np_merge(ak, av, self.ops)]
        if self.device == "cpu":
            self.table.clear()
        else:
            self.dropped_total += int(self.drops.item())
            self.tkeys.zero_()
            self.tvals.zero_()
            self.drops.zero_()
        return keys.shape[0]

    def _fold_np(self, keys, vals):
        """Rebucket key word 0 (relative time) to this table's coarser
        interval and merge."""
        iv = self.td.interval_s
        keys = keys.copy()
        keys[:, 0] = keys[:, 0] // iv * iv
        return _np_merge(keys, vals, self.ops)

    def _all_groups_np(self):
        """Merged (keys, vals) across the archive chunks + live."""
        if self.td.derive_from is not None:
            return self._fold_np(*self.source_table._all_groups_np())
        chunks = self.archive + [self._harvest_np()]
        keys = np.concatenate([k for k, _ in chunks], axis=0)
        vals = np.concatenate([v for _, v in chunks], axis=0)
        return _np_merge(keys, vals, self.ops)

    def rows(self) -> List[Dict]:
        keys, vals = self._all_groups_np()
        names = self.td.out_names or self.td.keys
        out = []
        for g in range(keys.shape[0]):
            row = {"time": self.time_base_s + int(keys[g, 0])}
            for ki, name in enumerate(names):
                row[name] = self._fmt_key(name, self.td.keys[ki],
                                          int(keys[g, 1 + ki]))
            row.update({f: int(v) for f, v in zip(self.fields, vals[g])})
            out.append(row)
        out.sort(key=lambda r: (r["time"],) + tuple(
            str(r[n]) for n in names))
        return out

    def drop_count(self) -> int:
        if self.td.derive_from is not None:
            return self.source_table.drop_count()
        if self.device == "cpu":
            return self.dropped_total
        return self.dropped_total + int(self.drops.item())

    # --------------------------------------------------------- checkpoint
    def state_dict(self):
        if self.td.derive_from is not None:
            return {"archive": list(self.archive)}
        if self.device == "cpu":
            return {"table": {k: list(v) for k, v in self.table.items()},
                    "archive": list(self.archive)}
        return {"tkeys": self.tkeys.cpu().clone(),
                "traw": self.traw.cpu().clone(),
                "tvals": self.tvals.cpu().clone(),
                "archive": list(self.archive)}

    def load_state_dict(self, st):
        self.archive = list(st.get("archive", []))
        if self.td.derive_from is not None:
            return
        if self.device == "cpu":
            if "table" in st:
                self.table.update(st["table"])
            return
        if "tkeys" in st:
            self.tkeys.copy_(st["tkeys"].to(self.tkeys.device))
            if "traw" in st:
                self.traw.copy_(st["traw"].to(self.traw.device))
            self.tvals.copy_(st["tvals"].to(self.tvals.device))


class RollupFamily:
    """All rollup tables for one pipeline source, updated per ingest
    batch (one kernel launch per table)."""

    def __init__(self, defs: List[TableDef], time_base_s: int,
                 device: str = "cpu"):
        self.tables: Dict[str, RollupTable] = {
            td.name: RollupTable(td, time_base_s, device, 1 << td.cap_pow2)
            for td in defs}
        for t in self.tables.values():
            if t.td.derive_from is not None:
                t.source_table = self.tables[t.td.derive_from]

    # archive-and-reset the live tables after this many rows: the
    # bounded device tables never saturate on long runs (the reference
    # flushes every closed interval to ClickHouse)
    FLUSH_EVERY_ROWS = 32_000_000

    def update(self, seg, base: int, n: int, stream: int = 0) -> None:
        tables = [t for t in self.tables.values()
                  if t.td.derive_from is None]
        if not tables or n == 0:
            return
        if tables[0].device == "cpu":
            for t in tables:
                t.update(seg, base, n, stream)
        else:
            from ..ops import gpu_ops
            gpu_ops.rollup_family(seg, base, n, tables[0].time_base_s,
                                  tables, stream=stream)
        self._rows_since_flush = getattr(self, "_rows_since_flush", 0) + n
        if self._rows_since_flush >= self.FLUSH_EVERY_ROWS:
            self.flush()

    def flush(self) -> None:
        """Interval flush: dump-and-reset every live table (one D2H
        copy each; derived tables fold at read and carry no state)."""
        for t in self.tables.values():
            t.flush_live()
        self._rows_since_flush = 0

    def get(self, name: str) -> Optional[RollupTable]:
        return self.tables.get(name)

    def state_dict(self):
        return {name: t.state_dict() for name, t in self.tables.items()}

    def load_state_dict(self, st):
        for name, s in st.items():
            if name in self.tables:
                self.tables[name].load_state_dict(s)
