"""Continuous-profiling ingest: Profile protobuf -> folded-stack store.

Reference counterpart: server/ingester/profile (decoder_parser.go folds and
compresses stacks into profile.in_process). Stacks are SmartEncoded: each
folded location string is interned once (id <-> string), rows store u32 ids.
"""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import Dict, List, Optional

from ..utils.stats import Counter
from ..wire import pb, metric, framing


@dataclass
class ProfileRow:
    timestamp: int          # us
    event_type: int
    location_id: int
    value: int              # count or duration (units per profile)
    pid: int = 0
    tid: int = 0
    pod_id: int = 0
    process_name: str = ""
    app_service: str = ""
    profile_language_type: str = ""


class ProfileStore:
    """Host-side columnar-ish store for profile samples + stack dictionary."""

    def __init__(self):
        self.rows: List[ProfileRow] = []
        self.loc_to_id: Dict[bytes, int] = {}
        self.id_to_loc: List[bytes] = []

    def intern(self, loc: bytes) -> int:
        i = self.loc_to_id.get(loc)
        if i is None:
            i = len(self.id_to_loc)
            self.loc_to_id[loc] = i
            self.id_to_loc.append(loc)
        return i

    def stored_bytes(self) -> int:
        return len(self.rows) * 40 + sum(len(s) for s in self.id_to_loc)


class ProfilePipeline:
    def __init__(self, counter: Optional[Counter] = None):
        self.store = ProfileStore()
        self.counter = counter or Counter("ingester.profile")

    def ingest_payload(self, payload: bytes) -> int:
        n = 0
        for rec in framing.iter_records(bytes(payload)):
            d = pb.decode(rec, metric.PROFILE)
            self.ingest_profile(d)
            n += 1
        self.counter.add("profiles_in", n)
        return n

    def ingest_profile(self, d: Dict) -> None:
        """One Profile message. `data` holds the folded stack (eBPF mode) or
        a pyroscope-format blob (external push; collapsed lines)."""
        data = d.get("data", b"")
        if d.get("data_compressed"):
            import ctypes as ct
            import numpy as np
            from ..ops import native
            lib = native.cpu()
            src = np.frombuffer(data, dtype=np.uint8)
            dst = np.zeros(max(len(data) * 20, 1 << 16), dtype=np.uint8)
            m = lib.df_zstd_decompress(src.ctypes.data, len(src),
                                       dst.ctypes.data, len(dst))
            data = dst[:m].tobytes() if m > 0 else b""
        fmt = d.get("format", "")
        common = dict(
            event_type=d.get("event_type", 0),
            pid=d.get("pid", 0), tid=d.get("tid", 0),
            pod_id=d.get("pod_id", 0),
            process_name=d.get("process_name", ""),
            app_service=d.get("name", ""),
            profile_language_type=d.get("spy_name", ""),
        )
        ts = d.get("timestamp") or (d.get("from_time", 0) * 1_000_000)
        if fmt == "folded" or "folded" in fmt or fmt == "":
            # one or more "stack;frames count" lines (collapsed format)
            count = d.get("count") or d.get("wide_count") or 1
            if b"\n" in data or b" " in data.strip():
                for line in data.splitlines():
                    line = line.strip()
                    if not line:
                        continue
                    stack, _, cnt = line.rpartition(b" ")
                    if stack and cnt.isdigit():
                        self._add(ts, stack, int(cnt), common)
                    else:
                        self._add(ts, line, count, common)
            elif data:
                self._add(ts, data, count, common)
        else:
            self._add(ts, data, d.get("count", 1), common)

    def _add(self, ts: int, stack: bytes, value: int, common: Dict) -> None:
        lid = self.store.intern(stack)
        self.store.rows.append(ProfileRow(
            timestamp=ts, location_id=lid, value=value, **common))


def build_flame(rows: List[ProfileRow], id_to_loc: List[bytes],
                event_type: Optional[int] = None,
                process_name: Optional[str] = None,
                time_start: int = 0, time_end: int = 1 << 62) -> Dict:
    """Merge folded stacks into a flame tree (reference:
    querier/profile/service/profile.go:84-330)."""
    root = {"name": "root", "value": 0, "self": 0, "children": {}}
    for r in rows:
        if event_type is not None and r.event_type != event_type:
            continue
        if process_name is not None and r.process_name != process_name:
            continue
        if not (time_start <= r.timestamp <= time_end):
            continue
        stack = id_to_loc[r.location_id].decode("utf-8", "replace")
        node = root
        root["value"] += r.value
        for frame in stack.split(";"):
            child = node["children"].get(frame)
            if child is None:
                child = {"name": frame, "value": 0, "self": 0, "children": {}}
                node["children"][frame] = child
            child["value"] += r.value
            node = child
        node["self"] += r.value

    def finalize(n):
        n["children"] = [finalize(c) for c in
                         sorted(n["children"].values(),
                                key=lambda c: -c["value"])]
        return n

    return finalize(root)


class ProfileApp:
    """HTTP surface: flame-graph query (reference /v1/profile/ProfileTracing)."""

    def __init__(self, pipeline: ProfilePipeline):
        self.pipe = pipeline

    def register(self, app) -> None:
        @app.get("/v1/profile/flame")
        def flame(process_name: str = None, event_type: int = None):
            st = self.pipe.store
            return build_flame(st.rows, st.id_to_loc,
                               event_type=event_type,
                               process_name=process_name)

        @app.get("/v1/profile/processes")
        def processes():
            return sorted({r.process_name for r in self.pipe.store.rows})
