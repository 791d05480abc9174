from .csr import LocalGraph, coo_to_csr, pad_feat_dim
from .synthetic import GlobalGraph, synth_graph, random_partitioned_graph, tiny_ring_graph, DATASET_SHAPES
from .ingest import load_graph_dir
from .partition import (range_assignment, bfs_assignment, grow_assignment,
                        edge_cut, build_local_graph,
                        partition_all, save_partitions, load_partition, global_degrees)

__all__ = [
    'LocalGraph', 'coo_to_csr', 'pad_feat_dim', 'GlobalGraph', 'synth_graph',
    'random_partitioned_graph', 'tiny_ring_graph', 'DATASET_SHAPES',
    'range_assignment', 'bfs_assignment', 'grow_assignment', 'edge_cut',
    'build_local_graph', 'partition_all',
    'save_partitions', 'load_partition', 'global_degrees', 'load_graph_dir',
]
