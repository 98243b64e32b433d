"""Prometheus remote-write pipeline: SmartEncoding for metrics.

The reference encodes every metric/label/value string to a global int ID
via the controller (prometheus encoder) and stores
ID-encoded rows in ClickHouse (prometheus_sample schema). Here the IDs
come from the same controller-global allocator (when wired) and the
sample columns live in torch tensors on the server's device — on a GPU
server the sample store is HBM-resident like the span store, and series
extraction is a device mask/gather instead of a Python scan.

Reference: server/ingester/prometheus/decoder (label encoding via
GetPrometheusLabelIDs), appendix D of SURVEY.md.
"""
from __future__ import annotations

from typing import Dict, List, Optional, Tuple

import numpy as np
import torch

from ..utils.stats import Counter
from ..wire import pb, prompb


class Interner:
    def __init__(self):
        self.to_id: Dict[str, int] = {}
        self.from_id: List[str] = []

    def name(self, i: int) -> str:
        return self.from_id[i] if 0 <= i < len(self.from_id) else ""

    def strings(self):
        return self.from_id

    def intern(self, s: str) -> int:
        i = self.to_id.get(s)
        if i is None:
            i = len(self.from_id)
            self.to_id[s] = i
            self.from_id.append(s)
        return i


class GlobalInterner(Interner):
    """Controller-backed: ids are cluster-global and persistent
    (GetPrometheusLabelIDs analog; round-1 weakness #48)."""

    def __init__(self, kind: str, alloc_fn):
        super().__init__()
        self.kind = kind
        self.alloc_fn = alloc_fn
        self.from_id_map: Dict[int, str] = {}

    def name(self, i: int) -> str:
        return self.from_id_map.get(i, "")

    def strings(self):
        return list(self.from_id_map.values())

    def intern(self, s: str) -> int:
        i = self.to_id.get(s)
        if i is None:
            i = self.alloc_fn({self.kind: [s]})[self.kind][s]
            self.to_id[s] = i
            self.from_id_map[i] = s
        return i


class _SampleColumns:
    """Append-only columnar sample store in torch tensors (series id,
    timestamp ms, value). Appends stage in host lists and flush into
    doubling device tensors — on a GPU server these columns are
    HBM-resident and matcher scans are device ops."""

    def __init__(self, device: str = "cpu"):
        self.device = torch.device(device)
        self.n = 0
        cap = 1 << 12
        self.series = torch.empty(cap, dtype=torch.int32,
                                  device=self.device)
        self.ts = torch.empty(cap, dtype=torch.int64, device=self.device)
        self.value = torch.empty(cap, dtype=torch.float64,
                                 device=self.device)
        self._st_series: List[int] = []
        self._st_ts: List[int] = []
        self._st_value: List[float] = []

    def append(self, sid: int, ts: int, value: float) -> None:
        self._st_series.append(sid)
        self._st_ts.append(ts)
        self._st_value.append(value)

    def _ensure(self, need: int) -> None:
        cap = self.series.numel()
        if need <= cap:
            return
        while cap < need:
            cap *= 2
        for name in ("series", "ts", "value"):
            old = getattr(self, name)
            new = torch.empty(cap, dtype=old.dtype, device=self.device)
            new[: self.n] = old[: self.n]
            setattr(self, name, new)

    def flush(self) -> None:
        k = len(self._st_series)
        if k == 0:
            return
        self._ensure(self.n + k)
        dev = self.device
        self.series[self.n: self.n + k] = torch.from_numpy(
            np.asarray(self._st_series, dtype=np.int32)).to(dev)
        self.ts[self.n: self.n + k] = torch.from_numpy(
            np.asarray(self._st_ts, dtype=np.int64)).to(dev)
        self.value[self.n: self.n + k] = torch.from_numpy(
            np.asarray(self._st_value, dtype=np.float64)).to(dev)
        self.n += k
        self._st_series.clear()
        self._st_ts.clear()
        self._st_value.clear()

    def for_series(self, sids: List[int]):
        """{sid: (ts_s ndarray, value ndarray)} via one device pass."""
        self.flush()
        if self.n == 0 or not sids:
            return {}
        col = self.series[: self.n]
        want = torch.tensor(sids, dtype=torch.int32, device=self.device)
        mask = torch.isin(col, want)
        idx = mask.nonzero(as_tuple=True)[0]
        sel_sid = col[idx].cpu().numpy()
        sel_ts = (self.ts[: self.n][idx] // 1000).cpu().numpy()
        sel_val = self.value[: self.n][idx].cpu().numpy()
        out = {}
        for sid in sids:
            m = sel_sid == sid
            out[sid] = (sel_ts[m], sel_val[m])
        return out

    def stored_bytes(self) -> int:
        self.flush()
        return self.n * (4 + 8 + 8)

    def counts_per_series(self) -> np.ndarray:
        self.flush()
        if self.n == 0:
            return np.zeros(0, dtype=np.int64)
        return np.bincount(self.series[: self.n].cpu().numpy())


class PromPipeline:
    def __init__(self, counter: Optional[Counter] = None,
                 id_allocator=None, device: str = "cpu"):
        """id_allocator: ControllerLite.alloc_prom_ids-shaped callable —
        when given, metric/label ids are controller-global + persistent
        instead of per-shard volatile (round-1 weakness #48)."""
        if id_allocator is not None:
            self.metric_names = GlobalInterner("metric", id_allocator)
            self.label_names = GlobalInterner("label_name", id_allocator)
            self.label_values = GlobalInterner("label_value", id_allocator)
        else:
            self.metric_names = Interner()
            self.label_names = Interner()
            self.label_values = Interner()
        # series: (metric_id, ((lname_id, lval_id), ...)) -> series_id
        self.series: Dict[Tuple, int] = {}
        self.series_labels: List[Tuple] = []
        self.series_metric: List[int] = []
        self.samples = _SampleColumns(device)
        self.counter = counter or Counter("ingester.prometheus")

    def _series_id(self, metric: str, lab_ids) -> int:
        mid = self.metric_names.intern(metric)
        key = (mid, tuple(sorted(lab_ids)))
        sid = self.series.get(key)
        if sid is None:
            sid = len(self.series_labels)
            self.series[key] = sid
            self.series_labels.append(key[1])
            self.series_metric.append(mid)
        return sid

    def ingest_labeled_samples(self, samples) -> int:
        """[(metric, labels_dict, ts_ms, value)] -> ID-encoded store.
        Entry point for OTLP metrics / telegraf (same SmartEncoding
        discipline as remote-write)."""
        n = 0
        for metric, labels, ts, value in samples:
            lab_ids = [(self.label_names.intern(k),
                        self.label_values.intern(str(v)))
                       for k, v in labels.items()]
            sid = self._series_id(metric, lab_ids)
            self.samples.append(sid, int(ts), float(value))
            n += 1
        self.counter.add("samples_in", n)
        return n

    def ingest_write_request(self, data: bytes) -> int:
        wr = pb.decode(data, prompb.WRITE_REQUEST)
        n = 0
        for ts in wr.get("timeseries", []):
            metric = ""
            lab_ids = []
            for lb in ts.get("labels", []):
                name = lb.get("name", "")
                value = lb.get("value", "")
                if name == "__name__":
                    metric = value
                    continue
                lab_ids.append((self.label_names.intern(name),
                                self.label_values.intern(value)))
            sid = self._series_id(metric, lab_ids)
            for sm in ts.get("samples", []):
                self.samples.append(sid, int(sm.get("timestamp", 0)),
                                    float(sm.get("value", 0.0)))
                n += 1
        self.counter.add("samples_in", n)
        return n

    # ------------------------------------------------------- query side
    def series_for(self, metric: str,
                   matchers: List[Tuple[str, str, str]] = ()) -> List[Dict]:
        """PromQL source: [{metric: labels, samples: {t_s: v}}]."""
        import re as _re
        mid = self.metric_names.to_id.get(metric)
        if mid is None:
            return []
        matched: List[Tuple[int, Dict[str, str]]] = []
        for sid, smid in enumerate(self.series_metric):
            if smid != mid:
                continue
            labels = {self.label_names.name(ln):
                      self.label_values.name(lv)
                      for ln, lv in self.series_labels[sid]}
            ok = True
            for lname, op, lval in matchers:
                got = labels.get(lname, "")
                if op == "=" and got != lval:
                    ok = False
                elif op == "!=" and got == lval:
                    ok = False
                elif op == "=~" and not _re.fullmatch(lval, got):
                    ok = False
                if not ok:
                    break
            if ok:
                matched.append((sid, labels))
        per_sid = self.samples.for_series([sid for sid, _ in matched])
        out = []
        for sid, labels in matched:
            ts_arr, val_arr = per_sid.get(sid, ((), ()))
            samples = {int(t): float(v) for t, v in zip(ts_arr, val_arr)}
            out.append({"metric": dict(labels, __name__=metric),
                        "samples": samples,
                        # raw scraped series are cumulative counters/gauges,
                        # unlike the engine's per-second rollup deltas
                        "kind": "counter"})
        return out

    def stored_bytes(self) -> int:
        """SmartEncoding accounting: ID-encoded samples + dictionaries."""
        samples = self.samples.stored_bytes()
        dicts = sum(len(s) for s in self.metric_names.strings()) + \
            sum(len(s) for s in self.label_names.strings()) + \
            sum(len(s) for s in self.label_values.strings())
        layout = sum(2 * 4 * len(t) for t in self.series_labels)
        return samples + dicts + layout

    def naive_bytes(self) -> int:
        """What the same samples cost with string labels per row."""
        counts = self.samples.counts_per_series()
        total = 0
        for sid, cnt in enumerate(counts):
            if cnt == 0:
                continue
            row = 8 + 8
            mid = self.series_metric[sid]
            row += len(self.metric_names.name(mid))
            for ln, lv in self.series_labels[sid]:
                row += len(self.label_names.name(ln)) + \
                    len(self.label_values.name(lv))
            total += row * int(cnt)
        return total

    def state_dict(self) -> dict:
        self.samples.flush()
        n = self.samples.n
        return {
            "interners": {
                "metric": list(self.metric_names.strings())
                if not isinstance(self.metric_names, GlobalInterner)
                else None,
                "to_id": {k: dict(v.to_id) for k, v in
                          (("metric", self.metric_names),
                           ("label_name", self.label_names),
                           ("label_value", self.label_values))},
            },
            "series": {k: v for k, v in self.series.items()},
            "series_labels": list(self.series_labels),
            "series_metric": list(self.series_metric),
            "samples": {
                "series": self.samples.series[:n].cpu().clone(),
                "ts": self.samples.ts[:n].cpu().clone(),
                "value": self.samples.value[:n].cpu().clone(),
            },
        }

    def load_state_dict(self, st: dict) -> None:
        for kind, itn in (("metric", self.metric_names),
                          ("label_name", self.label_names),
                          ("label_value", self.label_values)):
            for name, ident in st["interners"]["to_id"][kind].items():
                itn.to_id[name] = ident
                if isinstance(itn, GlobalInterner):
                    itn.from_id_map[ident] = name
                else:
                    while len(itn.from_id) <= ident:
                        itn.from_id.append("")
                    itn.from_id[ident] = name
        self.series = dict(st["series"])
        self.series_labels = list(st["series_labels"])
        self.series_metric = list(st["series_metric"])
        n = st["samples"]["series"].numel()
        self.samples._ensure(n)
        dev = self.samples.device
        self.samples.series[:n] = st["samples"]["series"].to(dev)
        self.samples.ts[:n] = st["samples"]["ts"].to(dev)
        self.samples.value[:n] = st["samples"]["value"].to(dev)
        self.samples.n = n
