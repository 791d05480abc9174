"""Property-based fuzz of the wire-layout planner and the partitioner
contract over random inputs."""
import torch
from hypothesis import given, settings, strategies as st

from adaqp_amd.comm.buffers import _layout, BITS_SET, bytes_per_node


@settings(max_examples=60, deadline=None)
@given(
    peers=st.lists(st.integers(0, 25), min_size=1, max_size=5),
    f=st.integers(1, 64),
    seed=st.integers(0, 10**6),
)
def test_sideplan_invariants_fuzz(peers, f, seed):
    gen = torch.Generator().manual_seed(seed)
    bits_pp, rows_pp = [], []
    base = 0
    for n in peers:
        if n == 0:
            bits_pp.append(None)
            rows_pp.append(None)
            continue
        bits_pp.append(torch.tensor(BITS_SET)[
            torch.randint(0, 3, (n,), generator=gen)])
        rows_pp.append(torch.arange(base, base + n))
        base += n
    plan = _layout(bits_pp, rows_pp, f)
    total = sum(peers)
    assert plan.total_nodes == total
    assert sum(plan.node_splits) == total
    assert sum(plan.byte_splits) == plan.total_bytes
    # pos is a permutation of [0, total)
    pos = torch.cat([plan.pos[b] for b in BITS_SET])
    assert torch.equal(torch.sort(pos).values, torch.arange(total))
    # byte ranges are disjoint and exactly cover the payload
    spans = []
    for b in BITS_SET:
        bpn = bytes_per_node(f, b)
        for o in plan.off[b].tolist():
            spans.append((o, o + bpn))
    spans.sort()
    cursor = 0
    for lo, hi in spans:
        assert lo == cursor, 'gap or overlap in wire layout'
        cursor = hi
    assert cursor == plan.total_bytes
    # rows align with pos ordering per bit group
    for b in BITS_SET:
        assert plan.rows[b].numel() == plan.pos[b].numel() == plan.off[b].numel()


@settings(max_examples=15, deadline=None)
@given(
    n=st.integers(40, 400),
    m=st.integers(100, 3000),
    p=st.integers(2, 6),
    seed=st.integers(0, 10**6),
)
def test_grow_contract_fuzz(n, m, p, seed):
    from adaqp_amd.graph import GlobalGraph, grow_assignment
    from adaqp_amd.graph.synthetic import _dedup_edges
    gen = torch.Generator().manual_seed(seed)
    s = torch.randint(0, n, (m,), generator=gen)
    d = torch.randint(0, n, (m,), generator=gen)
    s2 = torch.cat([s, d, torch.arange(n)])
    d2 = torch.cat([d, s, torch.arange(n)])
    s2, d2 = _dedup_edges(s2, d2, n)
    g = GlobalGraph(n, s2, d2, torch.zeros(n, 2),
                    torch.zeros(n, dtype=torch.int64),
                    torch.ones(n, dtype=torch.bool),
                    torch.zeros(n, dtype=torch.bool),
                    torch.zeros(n, dtype=torch.bool), 2, False)
    ga = grow_assignment(g, p, seed=seed % 97)
    assert ga.numel() == n
    assert int((ga < 0).sum()) == 0
    assert int(ga.max()) < p
    sizes = torch.bincount(ga, minlength=p)
    assert int(sizes.sum()) == n
    # balance within the documented tolerance (+ slack for tiny parts)
    assert float(sizes.max()) <= (n / p) * 1.10 + p + 1


@settings(max_examples=15, deadline=None)
@given(
    n=st.integers(30, 200),
    parts=st.integers(2, 4),
    seed=st.integers(0, 10**6),
    cut=st.floats(0.05, 0.6),
)
def test_exchange_structure_fuzz(n, parts, seed, cut):
    """For random partitioned graphs: every rank's LocalGraph validates,
    and the send/recv structure is globally consistent — rank r sends to
    q exactly the global ids q stores as r's remote block, in order."""
    from adaqp_amd.graph import random_partitioned_graph, partition_all
    g = random_partitioned_graph(n, 6 * n, 4, 3, parts, seed=seed,
                                 cut_frac=cut)
    lgs = partition_all(g, parts)
    assert sum(p.num_inner for p in lgs) == n
    for r, lg in enumerate(lgs):
        lg.validate()
        for q, other in enumerate(lgs):
            if q == r:
                continue
            # ids r sends to q (global, in r's send order)
            send_ids = (lg.local_to_global[lg.send_idx[q]]
                        if q in lg.send_idx else torch.empty(0, dtype=torch.int64))
            # q's remote block slice owned by r (global ids, stored order)
            base = other.num_inner + sum(other.recv_splits[:r])
            cnt = other.recv_splits[r]
            recv_ids = other.local_to_global[base:base + cnt]
            assert torch.equal(send_ids, recv_ids), (r, q)
