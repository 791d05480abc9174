"""Span timer with the reference's taxonomy (``AdaQP/util/timer.py``).

Two backends: wall-clock with stream-sync fences (reference-style, for
per-epoch breakdowns) and disabled (default — fences serialize the
overlap the framework exists to create, so production runs keep it off
and use rocprofv3 for kernel-level evidence).
"""
from __future__ import annotations

import time
from collections import defaultdict
from contextlib import contextmanager
from typing import Dict, List

import torch


class Timer:
    def __init__(self, enabled: bool = False, cuda: bool = False):
        self.enabled = enabled
        self.cuda = cuda
        self.records: Dict[str, float] = {}
        self.epoch_rows: List[List[float]] = []

    @contextmanager
    def record(self, name: str):
        if not self.enabled:
            yield
            return
        if self.cuda:
            torch.cuda.synchronize()
        t0 = time.perf_counter()
        yield
        if self.cuda:
            torch.cuda.synchronize()
        self.records[name] = self.records.get(name, 0.0) + time.perf_counter() - t0

    def clear(self):
        self.records = {}

    def epoch_rollup(self) -> List[float]:
        """[total_comm, quant+dequant, central_agg, marginal_agg, full_agg]
        (reference ``timer.py:29-51``)."""
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
