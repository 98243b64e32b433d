"""Deterministic sampling throttler (reference:
server/ingester/flow_log/throttler/throttling_queue.go:87-110).

Caps stored records per throttle window. The reference keeps a per-window
reservoir (math/rand) and flushes at window end; a streaming store can't
retract rows, so we sample with a hash threshold whose probability adapts to
the previous window's observed rate, with a hard per-window budget. Sampling
is seed-fixed (splitmix64 of (window, arrival position)) so golden tests are
reproducible (SURVEY.md §7 hard-part f).
"""
from __future__ import annotations

import numpy as np

MASK = (1 << 64) - 1


def _mix(x: int) -> int:
    x = (x ^ (x >> 30)) * 0xBF58476D1CE4E5B9 & MASK
    x = (x ^ (x >> 27)) * 0x94D049BB133111EB & MASK
    return (x ^ (x >> 31)) & MASK


class SamplingThrottler:
    def __init__(self, limit_per_window: int = 50000, window_s: int = 1,
                 seed: int = 0x5EED):
        self.limit = limit_per_window
        self.window_s = window_s
        self.seed = seed
        self.cur_window = -1
        self.seen_in_window = 0
        self.kept_in_window = 0
        self.prev_window_total = 0
        self.dropped = 0
        self.passed = 0

    def _roll_window(self, w: int) -> None:
        if w != self.cur_window:
            if self.cur_window >= 0:
                self.prev_window_total = self.seen_in_window
            self.cur_window = w
            self.seen_in_window = 0
            self.kept_in_window = 0

    def select(self, n: int, now_s: int) -> np.ndarray:
        """Sorted indices of records to keep from a batch of n arriving at
        now_s. Budget: at most `limit` kept per window."""
        w = now_s // self.window_s
        self._roll_window(w)
        start = self.seen_in_window
        self.seen_in_window += n
        budget = self.limit - self.kept_in_window
        if budget <= 0:
            self.dropped += n
            return np.empty(0, dtype=np.int64)
        # sampling probability from last window's rate (1.0 on a calm stream)
        est_rate = max(self.prev_window_total, start + n)
        p = min(1.0, self.limit / est_rate)
        if p >= 1.0 and n <= budget:
            self.kept_in_window += n
            self.passed += n
            return np.arange(n, dtype=np.int64)
        thresh = np.uint64(int(p * (2**64 - 1)))
        base = (self.seed * 0x9E3779B97F4A7C15 + w) & MASK
        keys = np.fromiter((_mix(base + start + i) for i in range(n)),
                           dtype=np.uint64, count=n)
        keep = np.nonzero(keys <= thresh)[0]
        if len(keep) > budget:
            keep = keep[:budget]
        self.kept_in_window += len(keep)
        self.passed += len(keep)
        self.dropped += n - len(keep)
        return keep.astype(np.int64)
