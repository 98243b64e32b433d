"""Continuous on-CPU profiler userspace: stack maps -> folded stacks.

Reference counterpart: agent/src/ebpf/user/profile/perf_profiler.c +
stringifier.c (folds kernel+user stacks into "a;b;c" strings, dedup by
stack-id pair). The kernel side here is progs.build_profiler (counts per
(tgid, ustack_id, kstack_id)); this module drains the maps, symbolizes
frames, and emits Profile dicts straight into the profile pipeline
(ingest/profile_pipeline.py — same store that serves the flame API).

Symbolization: /proc/<pid>/maps + ELF symtab walk, with a pluggable
resolver so tests (and the replay path) supply synthetic symbol tables.
"""
from __future__ import annotations

import struct
from typing import Callable, Dict, List, Optional, Tuple

COUNT_KEY_FMT = "<IiiI"   # tgid, ustack_id, kstack_id, pad


class DictSymbolizer:
    """Test/replay symbolizer: {(tgid, addr): name}."""

    def __init__(self, table: Dict[Tuple[int, int], str]):
        self.table = dict(table)

    def resolve(self, tgid: int, addr: int) -> str:
        return self.table.get((tgid, addr), f"0x{addr:x}")


class ProcSymbolizer:
    """Resolve addresses via /proc/<pid>/maps + ELF .symtab/.dynsym.

    Minimal self-contained ELF symbol reader (no external tooling);
    unknown frames keep their hex address (the reference stringifier
    does the same, stringifier.c)."""

    def __init__(self, proc_root: str = "/proc"):
        self.proc = proc_root
        self._maps: Dict[int, List[Tuple[int, int, int, str]]] = {}
        self._syms: Dict[str, List[Tuple[int, int, str]]] = {}

    def _load_maps(self, tgid: int):
        out = []
        try:
            with open(f"{self.proc}/{tgid}/maps") as f:
                for line in f:
                    parts = line.split()
                    if len(parts) < 6 or "x" not in parts[1]:
                        continue
                    lo, hi = (int(x, 16) for x in parts[0].split("-"))
                    off = int(parts[2], 16)
                    out.append((lo, hi, off, parts[5]))
        except OSError:
            pass
        self._maps[tgid] = out
        return out

    def _load_syms(self, path: str):
        if path in self._syms:
            return self._syms[path]
        syms: List[Tuple[int, int, str]] = []
        try:
            with open(path, "rb") as f:
                data = f.read()
            if data[:4] == b"\x7fELF" and data[4] == 2:
                shoff = struct.unpack_from("<Q", data, 0x28)[0]
                shentsize = struct.unpack_from("<H", data, 0x3A)[0]
                shnum = struct.unpack_from("<H", data, 0x3C)[0]
                secs = []
                for i in range(shnum):
                    base = shoff + i * shentsize
                    s_type = struct.unpack_from("<I", data, base + 4)[0]
                    s_off = struct.unpack_from("<Q", data, base + 0x18)[0]
                    s_size = struct.unpack_from("<Q", data, base + 0x20)[0]
                    s_link = struct.unpack_from("<I", data, base + 0x28)[0]
                    s_entsize = struct.unpack_from("<Q", data, base + 0x38)[0]
                    secs.append((s_type, s_off, s_size, s_link, s_entsize))
                for s_type, s_off, s_size, s_link, s_entsize in secs:
                    if s_type not in (2, 11) or not s_entsize:  # SYMTAB/DYNSYM
                        continue
                    str_off = secs[s_link][1]
                    for off in range(s_off, s_off + s_size, s_entsize):
                        name_i = struct.unpack_from("<I", data, off)[0]
                        info = data[off + 4]
                        value = struct.unpack_from("<Q", data, off + 8)[0]
                        size = struct.unpack_from("<Q", data, off + 16)[0]
                        if (info & 0xF) != 2 or not value:  # STT_FUNC
                            continue
                        end = data.index(b"\x00", str_off + name_i)
                        nm = data[str_off + name_i:end].decode(
                            "utf-8", "replace")
                        syms.append((value, size or 1, nm))
                syms.sort()
        except (OSError, ValueError, struct.error):
            pass
        self._syms[path] = syms
        return syms

    def resolve(self, tgid: int, addr: int) -> str:
        import bisect
        maps = self._maps.get(tgid) or self._load_maps(tgid)
        for lo, hi, off, path in maps:
            if lo <= addr < hi:
                syms = self._load_syms(path)
                file_addr = addr - lo + off
                for probe in (file_addr, addr):
                    i = bisect.bisect_right([s[0] for s in syms], probe) - 1
                    if 0 <= i < len(syms):
                        v, sz, nm = syms[i]
                        if v <= probe < v + max(sz, 1):
                            return nm
                return path.rsplit("/", 1)[-1]
        return f"0x{addr:x}"


def fold(counts: Dict[bytes, int], stacks: Dict[int, List[int]],
         symbolizer, comm_of: Optional[Callable[[int], str]] = None,
         ) -> List[Tuple[int, bytes, int]]:
    """(count-map entries, stack-id map) -> [(tgid, folded_stack, n)].

    Folded form is root-first "comm;frameN;...;frame0" like the
    reference stringifier; kernel frames get a "[k] " prefix."""
    out = []
    for key, n in counts.items():
        tgid, ustack, kstack, _ = struct.unpack(COUNT_KEY_FMT, key)
        frames: List[str] = []
        if kstack >= 0:
            for addr in stacks.get(kstack, []):
                frames.append("[k] " + symbolizer.resolve(0, addr))
        if ustack >= 0:
            for addr in stacks.get(ustack, []):
                frames.append(symbolizer.resolve(tgid, addr))
        frames.reverse()  # leaf-last (collapsed convention)
        comm = comm_of(tgid) if comm_of else f"pid-{tgid}"
        folded = ";".join([comm] + frames).encode()
        out.append((tgid, folded, n))
    return out


class CpuProfiler:
    """Drains the profiler maps on a cadence and ships folded stacks into
    the profile pipeline (event_type 0 = OnCPU)."""

    def __init__(self, pipeline, symbolizer=None, sample_freq: int = 99):
        self.pipeline = pipeline
        self.symbolizer = symbolizer or ProcSymbolizer()
        self.sample_freq = sample_freq

    def ingest_counts(self, counts: Dict[bytes, int],
                      stacks: Dict[int, List[int]], ts_ns: int,
                      comm_of=None) -> int:
        rows = fold(counts, stacks, self.symbolizer, comm_of)
        for tgid, folded, n in rows:
            self.pipeline.ingest_profile({
                "format": "folded", "data": folded, "count": n,
                "timestamp": ts_ns // 1000, "pid": tgid,
                "event_type": 0,
                "process_name": folded.split(b";", 1)[0].decode(),
                "name": "oncpu",
            })
        return len(rows)

    # live mode: drain kernel maps via loader fds
    def drain_kernel(self, map_fds: Dict[str, int], ts_ns: int) -> int:
        from . import loader
        counts: Dict[bytes, int] = {}
        key = None
        while True:
            key = loader.map_next_key(map_fds["counts"], key, 16)
            if key is None:
                break
            val = loader.map_lookup(map_fds["counts"], key, 8)
            if val is not None:
                counts[key] = struct.unpack("<Q", val)[0]
        stacks: Dict[int, List[int]] = {}
        for k in counts:
            for sid in struct.unpack(COUNT_KEY_FMT, k)[1:3]:
                if sid >= 0 and sid not in stacks:
                    raw = loader.map_lookup(map_fds["stacks"],
                                            struct.pack("<i", sid), 127 * 8)
                    if raw:
                        addrs = [a for (a,) in struct.iter_unpack("<Q", raw)
                                 if a]
                        stacks[sid] = addrs
        return self.ingest_counts(counts, stacks, ts_ns)
