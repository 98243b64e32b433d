"""DF-SQL subset parser -> query Plan.

Grammar (modeled on the reference querier's DF-SQL dialect,
server/querier/engine/clickhouse/clickhouse.go TransSelect/Where/GroupBy):

  SELECT item[, ...] FROM table [WHERE cond AND ...]
      [GROUP BY gitem[, ...]] [ORDER BY oitem [ASC|DESC][, ...]] [LIMIT n]
  item  := Func(metric) [AS alias] | tag [AS alias]
  Func  := Count | Sum | Avg | Max | Min
  cond  := tag op literal | time >= n | time <= n
  gitem := tag | time(seconds)
  SHOW tags|metrics FROM table  (handled by the engine)

String literals filter dict-encoded tags by compiling to SmartEncoding IDs
at plan time (the analog of the reference's dictGet/ID rewrite).
"""
from __future__ import annotations

import re
from typing import List, Optional, Tuple

from . import spec as Q
from .tags import L7_TAGS, L7_METRICS, TagDef

_TOKEN_RE = re.compile(
    r"\s*(?:(?P<num>\d+\.\d+|\d+)|(?P<str>'(?:[^'\\]|\\.)*'|\"(?:[^\"\\]|\\.)*\")"
    r"|(?P<id>`[^`]+`|[A-Za-z_][A-Za-z0-9_.]*)"
    r"|(?P<op><=|>=|!=|<>|=|<|>|\(|\)|,|\*))")

AGG_FUNCS = {"count": Q.AGGOP_COUNT, "sum": Q.AGGOP_SUM, "avg": None,
             "max": Q.AGGOP_MAX, "min": Q.AGGOP_MIN,
             "percentile": None, "apdex": None}


class SqlError(ValueError):
    pass


def tokenize(sql: str) -> List[Tuple[str, str]]:
    out = []
    pos = 0
    while pos < len(sql):
        m = _TOKEN_RE.match(sql, pos)
        if not m:
            if sql[pos:].strip() == "":
                break
            raise SqlError(f"bad token at: {sql[pos:pos+20]!r}")
        pos = m.end()
        if m.group("num"):
            out.append(("num", m.group("num")))
        elif m.group("str"):
            out.append(("str", m.group("str")[1:-1]))
        elif m.group("id"):
            ident = m.group("id")
            if ident.startswith("`"):
                ident = ident[1:-1]
            out.append(("id", ident))
        else:
            out.append(("op", m.group("op")))
    return out


class Parser:
    def __init__(self, tokens: List[Tuple[str, str]]):
        self.toks = tokens
        self.pos = 0

    def peek(self) -> Optional[Tuple[str, str]]:
        return self.toks[self.pos] if self.pos < len(self.toks) else None

    def next(self) -> Tuple[str, str]:
        t = self.peek()
        if t is None:
            raise SqlError("unexpected end of query")
        self.pos += 1
        return t

    def expect_kw(self, kw: str) -> None:
        t = self.next()
        if t[0] != "id" or t[1].lower() != kw:
            raise SqlError(f"expected {kw}, got {t[1]!r}")

    def kw_is(self, kw: str) -> bool:
        t = self.peek()
        return t is not None and t[0] == "id" and t[1].lower() == kw


def _resolve_tag(name: str, tags) -> TagDef:
    t = tags.get(name)
    if t is None:
        raise SqlError(f"unknown tag {name!r}")
    return t


def parse_sql(sql: str, dictionary=None, time_base_s: int = 0,
              tags=None, metrics=None, name_maps=None) -> Q.Plan:
    """Parse DF-SQL into a Plan against a table's tag map (default:
    l7_flow_log). `dictionary` compiles string literals on dict tags to
    SmartEncoding IDs (None -> impossible filters)."""
    tags = tags if tags is not None else L7_TAGS
    metrics = metrics if metrics is not None else L7_METRICS
    p = Parser(tokenize(sql))
    plan = Q.Plan(time_base_s=time_base_s)
    p.expect_kw("select")

    select_items: List[Tuple[str, Optional[str], Optional[str], str]] = []
    # (kind 'agg'|'tag', func, arg, alias)
    while True:
        t = p.next()
        alias = None
        if t[0] == "op" and t[1] == "*":
            select_items.append(("star", None, "*", "*"))
        elif t[0] == "id" and t[1].lower() in AGG_FUNCS and \
                p.peek() == ("op", "("):
            func = t[1].lower()
            p.next()  # (
            arg_t = p.next()
            arg = arg_t[1] if arg_t[1] != "*" else "*"
            param = None
            if p.peek() == ("op", ","):
                p.next()
                param = float(p.next()[1])
            nxt = p.next()
            if nxt != ("op", ")"):
                raise SqlError("expected )")
            alias = f"{func}({arg})"
            if p.kw_is("as"):
                p.next()
                alias = p.next()[1]
            select_items.append(("agg", func, (arg, param), alias))
        elif t[0] == "id" and t[1].lower() == "time" and p.peek() == ("op", "("):
            p.next()
            p.next()  # bucket value (the GROUP BY clause carries it)
            if p.next() != ("op", ")"):
                raise SqlError("expected )")
            alias = "time"
            if p.kw_is("as"):
                p.next()
                alias = p.next()[1]
            select_items.append(("tag", None, "time", alias))
        elif t[0] == "id":
            name = t[1]
            alias = name
            if p.kw_is("as"):
                p.next()
                alias = p.next()[1]
            select_items.append(("tag", None, name, alias))
        else:
            raise SqlError(f"bad select item {t!r}")
        if p.peek() == ("op", ","):
            p.next()
            continue
        break

    p.expect_kw("from")
    plan.table = p.next()[1]

    # WHERE: conjunction of simple terms, parenthesized OR-clauses and
    # IN lists (each OR/IN becomes a CNF group)
    if p.kw_is("where"):
        p.next()
        group_counter = 0
        while True:
            if p.peek() == ("op", "("):
                p.next()
                group_counter += 1
                while True:
                    name = p.next()
                    op_t = p.next()
                    lit = p.next()
                    _add_term(plan, name[1], op_t[1], lit, dictionary, tags,
                              group=group_counter, name_maps=name_maps)
                    if p.kw_is("or"):
                        p.next()
                        continue
                    if p.peek() == ("op", ")"):
                        p.next()
                        break
                    raise SqlError("expected OR or ) in clause")
            else:
                name = p.next()
                if name[0] != "id":
                    raise SqlError("expected tag in where")
                if p.kw_is("in"):
                    p.next()
                    if p.next() != ("op", "("):
                        raise SqlError("expected ( after IN")
                    group_counter += 1
                    while True:
                        lit = p.next()
                        _add_term(plan, name[1], "=", lit, dictionary, tags,
                                  group=group_counter,
                                  name_maps=name_maps)
                        if p.peek() == ("op", ","):
                            p.next()
                            continue
                        if p.next() == ("op", ")"):
                            break
                        raise SqlError("expected , or ) in IN list")
                else:
                    op_t = p.next()
                    if op_t[0] != "op" or op_t[1] not in Q.OP_BY_NAME:
                        raise SqlError(f"bad operator {op_t!r}")
                    lit = p.next()
                    _add_term(plan, name[1], op_t[1], lit, dictionary, tags,
                              name_maps=name_maps)
            if p.kw_is("and"):
                p.next()
                continue
            break

    # GROUP BY
    group_names: List[str] = []
    if p.kw_is("group"):
        p.next()
        p.expect_kw("by")
        while True:
            t = p.next()
            if t[0] != "id":
                raise SqlError("expected group key")
            if t[1].lower() == "time" and p.peek() == ("op", "("):
                p.next()
                bucket = int(p.next()[1])
                if p.next() != ("op", ")"):
                    raise SqlError("expected )")
                plan.keys.append(Q.Key(Q.SRC_TIME_BUCKET, 0, bucket))
                plan.key_names.append("time")
                plan.key_meta.append({"hydrate": "time"})
            elif t[1].lower() == "time":
                plan.keys.append(Q.Key(Q.SRC_TIME_BUCKET, 0, 1))
                plan.key_names.append("time")
                plan.key_meta.append({"hydrate": "time"})
            else:
                td = _resolve_tag(t[1], tags)
                plan.keys.append(Q.Key(td.family, td.idx))
                plan.key_names.append(t[1])
                plan.key_meta.append({"hydrate": td.hydrate})
            group_names.append(plan.key_names[-1])
            if p.peek() == ("op", ","):
                p.next()
                continue
            break

    # HAVING: post-aggregation filter on select aliases
    if p.kw_is("having"):
        p.next()
        having = []
        while True:
            name = p.next()[1]
            op_t = p.next()
            if op_t[0] != "op" or op_t[1] not in Q.OP_BY_NAME:
                raise SqlError(f"bad HAVING operator {op_t!r}")
            lit = p.next()
            if lit[0] != "num":
                raise SqlError("HAVING compares against a number")
            having.append((name, op_t[1], float(lit[1])))
            if p.kw_is("and"):
                p.next()
                continue
            break
        plan.having = having

    # ORDER BY / LIMIT
    if p.kw_is("order"):
        p.next()
        p.expect_kw("by")
        order = []
        while True:
            name = p.next()[1]
            desc = False
            if p.kw_is("desc"):
                p.next()
                desc = True
            elif p.kw_is("asc"):
                p.next()
            order.append((name, desc))
            if p.peek() == ("op", ","):
                p.next()
                continue
            break
        plan.order_by = order
    if p.kw_is("slimit"):
        # two-phase series limit: keep only the top-N groups by the first
        # aggregate (reference CHEngine QuerySlimitSql, clickhouse.go:627)
        p.next()
        plan.slimit = int(p.next()[1])
    if p.kw_is("limit"):
        p.next()
        plan.limit = int(p.next()[1])

    # build agg list / select list
    has_agg = any(k == "agg" for k, *_ in select_items)
    if has_agg or plan.keys:
        for kind, func, arg, alias in select_items:
            if kind == "tag":
                if arg not in group_names and arg != "time":
                    # selecting a non-grouped tag in agg query: treat as group
                    td = _resolve_tag(arg, tags)
                    plan.keys.append(Q.Key(td.family, td.idx))
                    plan.key_names.append(arg)
                    plan.key_meta.append({"hydrate": td.hydrate})
                continue
            if kind == "star":
                raise SqlError("SELECT * not valid in aggregated query")
            arg_name, arg_param = arg if isinstance(arg, tuple) else (arg, None)
            if func == "count":
                plan.aggs.append(Q.Agg(Q.AGGOP_COUNT))
                plan.agg_names.append(alias)
                plan.agg_meta.append({"op": "count"})
            elif func == "avg":
                md = metrics.get(arg_name) or _resolve_tag(arg_name, tags)
                plan.aggs.append(Q.Agg(Q.AGGOP_SUM, md.family, md.idx))
                plan.aggs.append(Q.Agg(Q.AGGOP_COUNT))
                plan.agg_names.append(alias)
                plan.agg_meta.append({"op": "avg"})
            elif func in ("percentile", "apdex"):
                # executed as a second pass over gathered values
                md = metrics.get(arg_name) or _resolve_tag(arg_name, tags)
                plan.aggs.append(Q.Agg(Q.AGGOP_COUNT))
                plan.agg_names.append(alias)
                plan.agg_meta.append({"op": func, "family": md.family,
                                      "idx": md.idx,
                                      "param": arg_param if arg_param
                                      is not None else
                                      (95 if func == "percentile" else
                                       100000)})
            else:
                md = metrics.get(arg_name) or _resolve_tag(arg_name, tags)
                op = AGG_FUNCS[func]
                plan.aggs.append(Q.Agg(op, md.family, md.idx))
                plan.agg_names.append(alias)
                plan.agg_meta.append({"op": func})
    else:
        plan.select_rows = True
        cols = []
        for kind, func, arg, alias in select_items:
            cols.append(arg)
        plan.select_cols = cols
    return plan


def _never(plan: Q.Plan, group: int) -> None:
    """A term that can never match (unknown dict string inside an OR
    group: the branch just contributes nothing)."""
    plan.terms.append(Q.Term(Q.SRC_CONST0, 0, Q.OP_EQ, 1, group=group))


def _add_term(plan: Q.Plan, name: str, op: str, lit, dictionary,
              tags, group: int = 0, name_maps=None) -> None:
    if name.startswith("attribute."):
        # custom-tag filter: both sides resolved to SmartEncoding ids
        # (reference: flow_tag custom_field_value filters)
        from ..store.l7_schema import DICT_DOM_ATTR_NAME, DICT_DOM_ATTR_VALUE
        if op not in ("=", "==", "!=", "<>"):
            raise SqlError("attribute.* supports = / != only")
        if lit[0] != "str":
            raise SqlError("attribute.* compares against a string")
        nid = dictionary.lookup_id(DICT_DOM_ATTR_NAME,
                                   name[len("attribute."):].encode()) \
            if dictionary is not None else None
        vid = dictionary.lookup_id(DICT_DOM_ATTR_VALUE, lit[1].encode()) \
            if dictionary is not None else None
        if nid is None or vid is None:
            if Q.OP_BY_NAME[op] == Q.OP_NE:
                return
            if group:
                _never(plan, group)
            else:
                plan.impossible = True
            return
        plan.terms.append(Q.Term(Q.SRC_ATTR_MATCH, 0, Q.OP_BY_NAME[op],
                                 nid, vid, group=group))
        return
    if name.lower() == "time":
        # time in epoch seconds against start_time (ns)
        v = int(lit[1]) * 10**9
        plan.terms.append(Q.Term(Q.SRC_U64, 0, Q.OP_BY_NAME[op], v,
                                 group=group))
        return
    td = _resolve_tag(name, tags)
    if lit[0] == "num":
        v = int(lit[1]) if "." not in lit[1] else int(float(lit[1]))
        plan.terms.append(Q.Term(td.family, td.idx, Q.OP_BY_NAME[op], v,
                                 group=group))
        return
    # string literal
    if td.hydrate == "raw":
        # row-table / CTE column: compare the Python value directly
        plan.terms.append(Q.Term(td.family, td.idx, Q.OP_BY_NAME[op],
                                 lit[1], group=group))
        return
    if td.hydrate == "strhash":
        from ..store.dictionary import str_hash_py
        v = str_hash_py(lit[1].encode(), Q.STR_FILTER_SEED)
        plan.terms.append(Q.Term(td.family, td.idx, Q.OP_BY_NAME[op], v,
                                 group=group))
        return
    if td.hydrate == "tracebin":
        # compile the literal exactly like the stored form: hex ids hash
        # as mix64(hi)^lo of the binary value, other strings as the
        # pooled-string hash (executor SRC_TRACE128 twin)
        from ..ops.ref import mix64
        from ..store.dictionary import str_hash_py
        lit_s = lit[1]
        hexok = len(lit_s) in (32, 16) and \
            all(c in "0123456789abcdefABCDEF" for c in lit_s)
        if hexok and len(lit_s) == 32 and td.idx == 0:
            v128 = int(lit_s, 16)
            v = mix64(v128 >> 64) ^ (v128 & ((1 << 64) - 1))
        elif hexok and len(lit_s) == 16 and td.idx == 1:
            v = int(lit_s, 16)
        else:
            v = str_hash_py(lit_s.encode(), Q.STR_FILTER_SEED)
        plan.terms.append(Q.Term(td.family, td.idx, Q.OP_BY_NAME[op], v,
                                 group=group))
        return
    if td.hydrate == "ip6str":
        # stored value is the packed 16-byte address
        import ipaddress
        from ..store.dictionary import str_hash_py
        try:
            packed = ipaddress.IPv6Address(lit[1]).packed
        except ValueError:
            raise SqlError(f"bad IPv6 literal {lit[1]!r}")
        v = str_hash_py(packed, Q.STR_FILTER_SEED)
        plan.terms.append(Q.Term(td.family, td.idx, Q.OP_BY_NAME[op], v,
                                 group=group))
        return
    if td.hydrate.startswith("dict:"):
        dom = int(td.hydrate.split(":")[1])
        ident = dictionary.lookup_id(dom, lit[1].encode()) \
            if dictionary is not None else None
        if ident is None:
            if Q.OP_BY_NAME[op] == Q.OP_NE:
                return  # != unknown-string matches everything
            if group:
                _never(plan, group)
            else:
                plan.impossible = True
            return
        plan.terms.append(Q.Term(td.family, td.idx, Q.OP_BY_NAME[op], ident,
                                 group=group))
    elif td.hydrate == "l7proto":
        from ..wire.const_enums import L7_PROTOCOL_NAMES
        rev = {v.lower(): k for k, v in L7_PROTOCOL_NAMES.items()}
        ident = rev.get(lit[1].lower())
        if ident is None:
            if group:
                _never(plan, group)
            else:
                plan.impossible = True
            return
        plan.terms.append(Q.Term(td.family, td.idx, Q.OP_BY_NAME[op], ident,
                                 group=group))
    elif td.hydrate.startswith("kgname:"):
        mp = td.hydrate.split(":", 1)[1]
        rev = {v: k for k, v in (name_maps or {}).get(mp, {}).items()}
        ident = rev.get(lit[1])
        if ident is None:
            if group:
                _never(plan, group)
            else:
                plan.impossible = True
            return
        plan.terms.append(Q.Term(td.family, td.idx, Q.OP_BY_NAME[op], ident,
                                 group=group))
    elif td.hydrate == "status":
        from .engine import STATUS_NAMES
        rev = {v.lower(): k for k, v in STATUS_NAMES.items()}
        ident = rev.get(lit[1].lower())
        if ident is None:
            if group:
                _never(plan, group)
            else:
                plan.impossible = True
            return
        plan.terms.append(Q.Term(td.family, td.idx, Q.OP_BY_NAME[op], ident,
                                 group=group))
    elif td.hydrate == "ip":
        import ipaddress
        v = int(ipaddress.IPv4Address(lit[1]))
        plan.terms.append(Q.Term(td.family, td.idx, Q.OP_BY_NAME[op], v,
                                 group=group))
    else:
        raise SqlError(f"tag {name} does not accept string literal")
