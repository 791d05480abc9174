#!/usr/bin/env python3
"""Overlap evidence bench: 2 ranks sharing one GPU via
the gloo-staged transport, reddit-shaped graph, per-mode epoch times.

The decomposed path enqueues the central SpMM on the default stream
BEFORE the host-blocking exchange (ops/dist_agg.py), so AdaQP-p should
beat Vanilla by (hidden exchange time); AdaQP-q shrinks the exchange
itself; AdaQP combines both. Run one mode under
``rocprofv3 --kernel-trace`` for the kernel-timeline evidence.

Usage:
    python tools/overlap_bench.py [--mode all|Vanilla|AdaQP|...]
        [--scale 0.25] [--epochs 10] [--out gpurun_out/overlap.json]
"""
import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), '..'))

MODES = ['Vanilla', 'AdaQP-p', 'AdaQP-q', 'AdaQP']


def worker(rank, world, port, mode, scale, dataset, epochs, warmup, dtype, q):
    os.environ.update(MASTER_ADDR='127.0.0.1', MASTER_PORT=str(port),
                      RANK=str(rank), WORLD_SIZE=str(world), LOCAL_RANK='0')
    import torch
    from adaqp_amd.comm import Communicator
    from adaqp_amd.runtime import GraphEngine
    from adaqp_amd.runtime.utils import train_epoch, global_train_count
    from adaqp_amd.models import DistGCN
    from adaqp_amd.helpers import RunMode, DistGNNType
    from adaqp_amd.graph import (synth_graph, range_assignment, global_degrees,
                                 build_local_graph, pad_feat_dim, DATASET_SHAPES)
    from adaqp_amd.ops.kernels import native
    comm = Communicator(backend='gloo')
    try:
        native()   # refuse to run the eager fallback silently
        dev = torch.device('cuda:0')
        torch.cuda.set_device(dev)
        comm.device = dev
        g = synth_graph(dataset, world, seed=17, scale=scale)
        assign = range_assignment(g.num_nodes, world)
        ind, outd = global_degrees(g)
        lg = build_local_graph(g, assign, rank, world, ind, outd)
        F = pad_feat_dim(lg, 8)
        C = DATASET_SHAPES[dataset][3]
        engine = GraphEngine(lg, RunMode(mode), DistGNNType.DistGCN,
                             msg_dims=[F, 256, 256], device=dev)
        if dtype == 'bf16':
            engine.compute_dtype = torch.bfloat16
        if engine.bit_type.name == 'QUANT':
            engine.set_uniform_assignment(4)
        torch.manual_seed(33)
        model = DistGCN(F, 256, C, num_layers=3).to(dev)
        comm.sync_model_params(model)
        opt = torch.optim.Adam(model.parameters(), lr=0.01)
        gc = global_train_count(engine)
        for _ in range(warmup):
            train_epoch(engine, model, opt, gc, False)
        comm.barrier()
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(epochs):
            train_epoch(engine, model, opt, gc, False)
        torch.cuda.synchronize()
        el = torch.tensor([time.perf_counter() - t0])
        comm.all_reduce_max(el)
        if rank == 0:
            ms = float(el.item()) / epochs * 1e3
            if q is None:
                print(f'{mode}: {ms:.1f} ms/epoch', flush=True)
            else:
                q.put((mode, ms))
    finally:
        Communicator.shutdown()


def main():
    import torch.multiprocessing as mp
    p = argparse.ArgumentParser()
    p.add_argument('--mode', default='all')
    p.add_argument('--dataset', default='reddit')
    p.add_argument('--scale', type=float, default=0.25)
    p.add_argument('--epochs', type=int, default=10)
    p.add_argument('--warmup', type=int, default=3)
    p.add_argument('--dtype', default='fp32', choices=['fp32', 'bf16'])
    p.add_argument('--out', default=None)
    p.add_argument('--worker', type=int, default=None,
                   help='run ONE rank in-process (no mp.spawn) so a '
                        'profiler can wrap a single worker; launch the '
                        'peer rank separately with the other index')
    p.add_argument('--port', type=int, default=29640)
    args = p.parse_args()
    if args.worker is not None:
        mode = args.mode if args.mode != 'all' else 'AdaQP-p'
        worker(args.worker, 2, args.port, mode, args.scale, args.dataset,
               args.epochs, args.warmup, args.dtype, None)
        return
    modes = MODES if args.mode == 'all' else [args.mode]
    ctx = mp.get_context('spawn')
    results = {}
    for i, mode in enumerate(modes):
        q = ctx.SimpleQueue()
        procs = [ctx.Process(target=worker,
                             args=(r, 2, 29610 + i, mode, args.scale,
                                   args.dataset, args.epochs, args.warmup,
                                   args.dtype, q))
                 for r in range(2)]
        for pr in procs:
            pr.start()
        for pr in procs:
            pr.join(900)
            if pr.is_alive():
                pr.terminate()
                raise SystemExit(f'{mode}: hung')
            if pr.exitcode != 0:
                raise SystemExit(f'{mode}: worker failed rc={pr.exitcode}')
        m, ms = q.get()
        results[m] = ms
        print(f'{m}: {ms:.1f} ms/epoch', flush=True)
    if args.out:
        os.makedirs(os.path.dirname(args.out) or '.', exist_ok=True)
        json.dump({'dataset': args.dataset, 'scale': args.scale,
                   'world': 2, 'dtype': args.dtype,
                   'ms_per_epoch': results}, open(args.out, 'w'), indent=1)


if __name__ == '__main__':
    main()
