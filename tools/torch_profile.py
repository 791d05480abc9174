#!/usr/bin/env python3
"""Op-level attribution of one flagship epoch via torch.profiler
(kernel stats say WHICH kernels run; this says which PYTHON OPS launch
them — e.g. the reduce_kernel share rocprof can't attribute).

Usage: python tools/torch_profile.py [--scale 0.3] [--dtype fp32]
"""
import argparse
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), '..'))
import torch


def main():
    p = argparse.ArgumentParser()
    p.add_argument('--scale', type=float, default=0.3)
    p.add_argument('--dtype', default='fp32', choices=['fp32', 'bf16'])
    p.add_argument('--dataset', default='ogbn-products')
    args = p.parse_args()
    os.environ.setdefault('MASTER_ADDR', '127.0.0.1')
    os.environ.setdefault('MASTER_PORT', '29655')
    os.environ.setdefault('RANK', '0')
    os.environ.setdefault('WORLD_SIZE', '1')
    os.environ.setdefault('LOCAL_RANK', '0')

    from adaqp_amd.comm import Communicator
    from adaqp_amd.runtime import GraphEngine
    from adaqp_amd.runtime.utils import train_epoch, global_train_count
    from adaqp_amd.models import DistSAGE
    from adaqp_amd.helpers import RunMode, DistGNNType
    from adaqp_amd.graph import (synth_graph, range_assignment, global_degrees,
                                 build_local_graph, pad_feat_dim, DATASET_SHAPES)
    comm = Communicator()
    g = synth_graph(args.dataset, 1, seed=17, scale=args.scale)
    assign = range_assignment(g.num_nodes, 1)
    ind, outd = global_degrees(g)
    lg = build_local_graph(g, assign, 0, 1, ind, outd)
    F = pad_feat_dim(lg, 8)
    C = DATASET_SHAPES[args.dataset][3]
    engine = GraphEngine(lg, RunMode('AdaQP'), DistGNNType.DistSAGE,
                         msg_dims=[F, 256, 256], device=comm.device)
    if args.dtype == 'bf16':
        engine.compute_dtype = torch.bfloat16
    engine.set_uniform_assignment(4)
    model = DistSAGE(F, 256, C, 3).to(comm.device)
    opt = torch.optim.Adam(model.parameters(), lr=0.01)
    gc = global_train_count(engine)
    for _ in range(3):
        train_epoch(engine, model, opt, gc, False)
    torch.cuda.synchronize()
    from torch.profiler import profile, ProfilerActivity
    with profile(activities=[ProfilerActivity.CPU, ProfilerActivity.CUDA]) as prof:
        for _ in range(2):
            train_epoch(engine, model, opt, gc, False)
        torch.cuda.synchronize()
    print(prof.key_averages().table(sort_by='self_cuda_time_total',
                                    row_limit=30))
    Communicator.shutdown()


if __name__ == '__main__':
    main()
