"""Assigner unit tests: MILP behavior with fabricated matrices (no comm)."""
import numpy as np
import pytest
import torch

from adaqp_amd.assigner.assigner import Assigner, BITS_COST
from adaqp_amd.comm.buffers import BITS_SET


def _bare_assigner(lam):
    a = object.__new__(Assigner)
    a.coe_lambda = lam
    a.init_bits = 8
    a.solver_time_limit = 10.0
    a.group_size = 2
    return a


def _mk_groups(G, F=64, count=2, var_scale=1.0):
    var = np.zeros((G, len(BITS_SET)))
    mb = np.zeros((G, len(BITS_SET)))
    for gi in range(G):
        s = var_scale * (G - gi)   # earlier groups = higher variance
        for bi, b in enumerate(BITS_SET):
            var[gi, bi] = BITS_COST[b] * s
            mb[gi, bi] = count * (F * b / 8 + 4) / 2 ** 20
    return {'counts': [count] * G, 'var': var, 'mb': mb}


def test_milp_extremes():
    cost = {(0, 1): (1.0, 1e-4), (1, 0): (1.0, 1e-4)}
    rank_groups = [{1: _mk_groups(3)}, {0: _mk_groups(3)}]
    # lambda=1: pure variance objective -> max bits everywhere
    sol = _bare_assigner(1.0)._solve_key('forward0', rank_groups, cost)
    assert all(all(b == len(BITS_SET) - 1 for b in ch)
               for r in sol for ch in r.values())
    # lambda=0: pure time objective -> min bits everywhere
    sol = _bare_assigner(0.0)._solve_key('forward0', rank_groups, cost)
    assert all(all(b == 0 for b in ch) for r in sol for ch in r.values())


def test_milp_mixed_assignment():
    """Intermediate lambda with a slow channel: the slow channel should get
    lower bits than the fast one for equal variance."""
    cost = {(0, 1): (100.0, 1e-4),   # slow link
            (1, 0): (0.001, 1e-4)}   # fast link
    rank_groups = [{1: _mk_groups(4, var_scale=1.0)},
                   {0: _mk_groups(4, var_scale=1.0)}]
    sol = _bare_assigner(0.5)._solve_key('forward0', rank_groups, cost)
    slow_bits = sum(sol[0][1])
    fast_bits = sum(sol[1][0])
    assert fast_bits >= slow_bits
    # fast link has no reason not to use 8 bits (bit index 2) everywhere
    assert fast_bits == 2 * 4


def test_milp_empty():
    sol = _bare_assigner(0.5)._solve_key('forward0', [{}, {}], {})
    assert sol == [{}, {}]
