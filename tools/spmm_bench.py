#!/usr/bin/env python3
"""Self-contained SpMM microbench (no partition machinery, safe under
rocprofv3). Builds a products-like CSR directly on the GPU and times the
spmm_csr kernel; prints effective gather TB/s."""
import argparse
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), '..'))

import torch


def build_csr(rows, deg_mean, n_cols, device, seed=0, hub_frac=0.001):
    g = torch.Generator(device='cpu').manual_seed(seed)
    deg = torch.full((rows,), deg_mean, dtype=torch.int64)
    nhub = max(int(rows * hub_frac), 1)
    deg[torch.randperm(rows, generator=g)[:nhub]] = deg_mean * 100
    indptr = torch.zeros(rows + 1, dtype=torch.int64)
    indptr[1:] = torch.cumsum(deg, 0)
    E = int(indptr[-1])
    # clustered columns: around the row's own neighborhood
    r = torch.repeat_interleave(torch.arange(rows), deg)
    off = (torch.randn(E, generator=g) * (n_cols * 0.01)).long()
    idx = (r * (n_cols // max(rows, 1)) + off).clamp_(0, n_cols - 1)
    return indptr.to(device), idx.to(device), E


def main():
    p = argparse.ArgumentParser()
    p.add_argument('--rows', type=int, default=700_000)
    p.add_argument('--cols', type=int, default=750_000)
    p.add_argument('--deg', type=int, default=50)
    p.add_argument('--F', type=int, default=256)
    p.add_argument('--iters', type=int, default=10)
    p.add_argument('--col-block', type=int, default=0)
    args = p.parse_args()
    assert torch.cuda.is_available()
    from adaqp_amd.ops.kernels import SpmmView, spmm, native
    native()
    dev = torch.device('cuda')
    indptr, indices, E = build_csr(args.rows, args.deg, args.cols, dev)
    view = SpmmView(indptr, indices, 0, args.rows,
                    col_block=args.col_block).to(dev)
    x = torch.randn(args.cols, args.F, device=dev)
    src = torch.rand(args.cols, device=dev) + 0.5
    dst = torch.rand(args.rows, device=dev) + 0.5
    for _ in range(3):
        y = spmm(view, x, None, src, dst)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.iters):
        y = spmm(view, x, None, src, dst)
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / args.iters
    gather_bytes = E * (args.F * 4 + 12)
    print(f'rows={args.rows} E={E} F={args.F} cb={args.col_block}: {dt*1e3:.3f} ms/call, '
          f'apparent gather {gather_bytes/dt/1e12:.2f} TB/s')
    sys.stdout.flush()


if __name__ == '__main__':
    main()
