"""QuerySpec ABI — ctypes mirror of dfgpu.hip's QTerm/QKey/QAgg/QuerySpec.

tests/test_query_spec.py validates sizeof against df_spec_sizes() so the two
sides cannot drift silently.
"""
from __future__ import annotations

import ctypes as ct
from dataclasses import dataclass, field
from typing import List, Optional

QMAX_TERMS = 16
QMAX_KEYS = 4
QMAX_AGGS = 8

# source families (dfgpu.hip enum)
SRC_U64 = 0
SRC_U32 = 1
SRC_U8 = 2
SRC_DID = 3
SRC_KG = 4
SRC_ATTR_VAL = 5
SRC_TIME_BUCKET = 6
SRC_CONST0 = 7
SRC_STR_HASH = 8
SRC_ATTR_MATCH = 9  # filter-only: attr name_id == v0 AND value_id == v1
# binary-transcoded OTel ids: idx 0 = trace_id (mix64(hi)^lo), 1 = span_id;
# zero binary cols fall back to the pooled-string hash (non-hex ids)
SRC_TRACE128 = 10

# seed for pooled-string filter hashing (twin: dfgpu.hip STR_FILTER_SEED)
STR_FILTER_SEED = 0x5157A15E5EED

OP_EQ, OP_NE, OP_LT, OP_LE, OP_GT, OP_GE, OP_BETWEEN = range(7)
AGGOP_COUNT, AGGOP_SUM, AGGOP_MIN, AGGOP_MAX = range(4)

OP_BY_NAME = {"=": OP_EQ, "==": OP_EQ, "!=": OP_NE, "<>": OP_NE, "<": OP_LT,
              "<=": OP_LE, ">": OP_GT, ">=": OP_GE}


class QTermC(ct.Structure):
    _fields_ = [("family", ct.c_uint8), ("op", ct.c_uint8),
                ("idx", ct.c_uint16), ("group", ct.c_uint8),
                ("v0", ct.c_uint64), ("v1", ct.c_uint64)]


class QKeyC(ct.Structure):
    _fields_ = [("family", ct.c_uint8), ("idx", ct.c_uint16),
                ("bucket", ct.c_uint32)]


class QAggC(ct.Structure):
    _fields_ = [("op", ct.c_uint8), ("family", ct.c_uint8),
                ("idx", ct.c_uint16)]


class QuerySpecC(ct.Structure):
    _fields_ = [("terms", QTermC * QMAX_TERMS),
                ("keys", QKeyC * QMAX_KEYS),
                ("aggs", QAggC * QMAX_AGGS),
                ("n_terms", ct.c_uint32), ("n_keys", ct.c_uint32),
                ("n_aggs", ct.c_uint32),
                ("time_base_s", ct.c_uint64)]


@dataclass
class Term:
    family: int
    idx: int
    op: int
    v0: int
    v1: int = 0
    group: int = 0  # 0 = AND; >=1 = member of that OR-clause


@dataclass
class Key:
    family: int
    idx: int
    bucket: int = 0  # seconds per bucket for SRC_TIME_BUCKET


@dataclass
class Agg:
    op: int
    family: int = SRC_CONST0
    idx: int = 0


@dataclass
class Plan:
    table: str = "l7_flow_log"
    terms: List[Term] = field(default_factory=list)
    keys: List[Key] = field(default_factory=list)
    aggs: List[Agg] = field(default_factory=list)
    time_base_s: int = 0
    # host-side metadata (not part of the kernel spec)
    key_names: List[str] = field(default_factory=list)
    agg_names: List[str] = field(default_factory=list)
    key_meta: List[dict] = field(default_factory=list)  # hydration info
    agg_meta: List[dict] = field(default_factory=list)
    order_by: Optional[List] = None
    having: Optional[List] = None  # [(column, op, number)] post-agg
    limit: Optional[int] = None
    slimit: Optional[int] = None
    select_rows: bool = False  # non-aggregated SELECT
    select_cols: List[str] = field(default_factory=list)
    impossible: bool = False   # filter references unknown dict string

    def to_bytes(self) -> bytes:
        c = QuerySpecC()
        assert len(self.terms) <= QMAX_TERMS
        assert len(self.keys) <= QMAX_KEYS
        assert len(self.aggs) <= QMAX_AGGS
        for i, t in enumerate(self.terms):
            c.terms[i] = QTermC(family=t.family, op=t.op, idx=t.idx,
                                group=getattr(t, "group", 0),
                                v0=t.v0 & (2**64 - 1), v1=t.v1 & (2**64 - 1))
        for i, k in enumerate(self.keys):
            c.keys[i] = QKeyC(family=k.family, idx=k.idx, bucket=k.bucket)
        for i, a in enumerate(self.aggs):
            c.aggs[i] = QAggC(op=a.op, family=a.family, idx=a.idx)
        c.n_terms, c.n_keys, c.n_aggs = (len(self.terms), len(self.keys),
                                         len(self.aggs))
        c.time_base_s = self.time_base_s
        return bytes(memoryview(c))
