"""Span timer with the reference's taxonomy (``AdaQP/util/timer.py``).

Three backends:
- disabled (default): production runs keep timing off and use rocprofv3.
- ``mode='sync'``: wall clock with stream-sync fences (reference-style,
  ``timer.py:18-27``) — accurate totals but SERIALIZES the comm/compute
  overlap it measures.
- ``mode='events'``: hipEvent pairs recorded on the current stream; spans
  are resolved at epoch rollup with one sync — measures GPU span time
  without perturbing the overlap (the MI355X-native replacement the
  survey calls for).
"""
from __future__ import annotations

import time
from collections import defaultdict
from contextlib import contextmanager
from typing import Dict, List, Tuple

import torch


class Timer:
    def __init__(self, enabled: bool = False, cuda: bool = False,
                 mode: str = 'sync'):
        self.enabled = enabled
        self.cuda = cuda
        self.mode = mode if cuda else 'sync'
        self.records: Dict[str, float] = {}
        self.pending: List[Tuple[str, torch.cuda.Event, torch.cuda.Event]] = []
        self.epoch_rows: List[List[float]] = []

    @contextmanager
    def record(self, name: str):
        if not self.enabled:
            yield
            return
        if self.cuda and self.mode == 'events':
            e0 = torch.cuda.Event(enable_timing=True)
            e1 = torch.cuda.Event(enable_timing=True)
            e0.record()
            yield
            e1.record()
            self.pending.append((name, e0, e1))
            return
        if self.cuda:
            torch.cuda.synchronize()
        t0 = time.perf_counter()
        yield
        if self.cuda:
            torch.cuda.synchronize()
        self.records[name] = self.records.get(name, 0.0) + time.perf_counter() - t0

    def _drain_events(self):
        if not self.pending:
            return
        torch.cuda.synchronize()
        for name, e0, e1 in self.pending:
            self.records[name] = (self.records.get(name, 0.0)
                                  + e0.elapsed_time(e1) / 1e3)
        self.pending = []

    def clear(self):
        self.records = {}
        self.pending = []

    def epoch_rollup(self) -> List[float]:
        """[comm, quant+dequant, central_agg, marginal_agg, full_agg,
        grad_reduce] (reference ``timer.py:29-51``).

        SEMANTICS (events mode): each value is the summed GPU *span* time
        of that bucket, measured per-stream without serializing. Spans on
        different streams can OVERLAP in wall time — by design, the comm
        bucket (comm stream) runs concurrently with central_agg (default
        stream) in the decomposed path — so the row may sum to MORE than
        the epoch's wall time. Read it as span accounting, not as a
        partition of the epoch: wall saved by overlap shows up as
        (sum of buckets) − (epoch wall time). In 'sync' mode fences
        serialize every span, so the row does partition wall time but the
        overlap being measured is destroyed (reference behavior)."""
        if self.cuda and self.mode == 'events':
            self._drain_events()
        buckets = defaultdict(float)
        for k, v in self.records.items():
            if 'exchange' in k:
                buckets['comm'] += v
            elif 'quant' in k:
                buckets['quant'] += v
            elif 'central' in k:
                buckets['central'] += v
            elif 'marginal' in k:
                buckets['marginal'] += v
            elif 'full' in k:
                buckets['full'] += v
            elif 'reduce' in k:
                buckets['reduce'] += v
        row = [buckets['comm'], buckets['quant'], buckets['central'],
               buckets['marginal'], buckets['full'], buckets['reduce']]
        self.epoch_rows.append(row)
        self.clear()
        return row
