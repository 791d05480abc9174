#!/usr/bin/env python3
"""Offline graph partitioning CLI.

Same surface as the reference (``/root/reference/graph_partition.py:5-16``:
dataset, raw_dir, partition_dir, partition_size). There is no network in
this environment, so datasets are SYNTHETIC graphs of the named shape
(adaqp_amd/graph/synthetic.py); partitioning is range (planted locality)
or greedy BFS.
"""
import argparse

from adaqp_amd.graph import synth_graph, partition_all, save_partitions


def main():
    p = argparse.ArgumentParser()
    p.add_argument('--dataset', type=str, default='reddit',
                   choices=['reddit', 'yelp', 'ogbn-products', 'amazonProducts'])
    p.add_argument('--raw_dir', type=str, default=None,
                   help='load a REAL dataset from this GraphSAINT-layout '
                        'directory (adj_full.npz + feats.npy + class_map/'
                        'labels + role.json) instead of generating a '
                        'synthetic graph (reference --raw_dir parity)')
    p.add_argument('--partition_dir', type=str, default='part_data')
    p.add_argument('--partition_size', type=int, default=4)
    p.add_argument('--method', type=str, default='range',
                   choices=['range', 'grow', 'bfs'],
                   help="'range' exploits the synthetic generator's planted "
                        "locality (METIS-quality there, O(1)); 'grow' is the "
                        "vectorized balanced region-growing partitioner for "
                        "arbitrary graphs ('bfs' is its back-compat alias)")
    p.add_argument('--cut_frac', type=float, default=0.10)
    p.add_argument('--scale', type=float, default=1.0)
    p.add_argument('--seed', type=int, default=17)
    args = p.parse_args()

    if args.raw_dir:
        from adaqp_amd.graph import load_graph_dir
        g = load_graph_dir(args.raw_dir)
        if args.method == 'range':
            # contiguous ranges carry no locality on arbitrary data
            args.method = 'grow'
    else:
        g = synth_graph(args.dataset, args.partition_size, seed=args.seed,
                        cut_frac=args.cut_frac, scale=args.scale)
    parts = partition_all(g, args.partition_size, method=args.method)
    d = save_partitions(parts, args.partition_dir, args.dataset,
                        meta={'method': args.method, 'cut_frac': args.cut_frac,
                              'scale': args.scale, 'seed': args.seed})
    cut = sum(p.num_remote for p in parts)
    print(f'saved {args.partition_size} partitions to {d} '
          f'({g.num_nodes} nodes, {g.num_edges} edges, '
          f'{cut} halo nodes total)')


if __name__ == '__main__':
    main()
