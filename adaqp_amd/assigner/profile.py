"""Per-channel α-β communication cost model.

Reference parity: ``AdaQP/assigner/profile.py`` (timed dummy p2p sweeps,
deg-1 polyfit time = α·MB + β per directed pair). Redesigned for xGMI:
links are point-to-point and concurrent, so the MILP models epoch
exchange time as max over directed channels (not the reference's gloo
ring rounds); the per-channel α-β fit is what feeds it. Sizes span KBs
to tens of MBs (xGMI ≈153 GB/s per link — β dominates small messages).
"""
from __future__ import annotations

import time
from typing import Dict, List, Tuple

import numpy as np
import torch
import torch.distributed as dist

from ..comm.communicator import Communicator


def _sizes(num_points: int, max_mb: float = 16.0) -> List[int]:
    return [int(s) for s in np.logspace(np.log10(4096), np.log10(max_mb * 2 ** 20),
                                        num_points)]


def fit_cost_models(comm: Communicator, num_points: int = 12,
                    iters: int = 3) -> Dict[Tuple[int, int], Tuple[float, float]]:
    """Returns {(src, dst): (alpha_s_per_MB, beta_s)} for all directed pairs.

    Pairs are profiled one at a time (sender-by-sender, reference
    ``profile.py:46-83``) so each measurement sees an idle fabric.
    """
    W, rank = comm.world_size, comm.rank
    models: Dict[Tuple[int, int], Tuple[float, float]] = {}
    if W == 1:
        return models
    dev = comm.device
    # gloo cannot carry CUDA tensors over p2p; in the gloo-staged debug
    # transport the real exchange stages via host memory anyway, so the
    # cost model should measure host-side gloo too.
    if dev.type == 'cuda' and 'nccl' not in comm.backend:
        dev = torch.device('cpu')
    sizes = _sizes(num_points)
    buf = torch.empty(sizes[-1], dtype=torch.uint8, device=dev)
    mins: Dict[Tuple[int, int], List[float]] = {}
    for s in range(W):
        for r in range(W):
            if s == r:
                continue
            comm.barrier()
            if rank not in (s, r):
                continue
            per_size = []
            for size in sizes:
                t = buf[:size]
                best = float('inf')
                for it in range(iters + 1):
                    if dev.type == 'cuda':
                        torch.cuda.synchronize()
                    t0 = time.perf_counter()
                    if rank == s:
                        dist.send(t, dst=r)
                    else:
                        dist.recv(t, src=s)
                    if dev.type == 'cuda':
                        torch.cuda.synchronize()
                    if it > 0:   # first transfer is warmup
                        best = min(best, time.perf_counter() - t0)
                per_size.append(best)
            mins[(s, r)] = per_size
    comm.barrier()
    # the receiver of each channel fits it; models are all-gathered
    local = {}
    for (s, r), per_size in mins.items():
        if rank != r:
            continue
        mb = np.array(sizes) / 2 ** 20
        a, b = np.polyfit(mb, np.array(per_size), 1)
        local[(s, r)] = (max(float(a), 1e-9), max(float(b), 0.0))
    for d in comm.all_gather_object(local):
        models.update(d)
    return models
