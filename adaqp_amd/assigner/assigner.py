"""Adaptive bit-width assigner.

Reference parity: ``AdaQP/assigner/assigner.py`` — three schemes
(uniform / random / adaptive), variance tracing, score-weighted
grouping, rank-0 MILP solve + scatter, group->node expansion.

MI355X redesign:
- solver is scipy's HiGHS MILP (open; the reference wants Gurobi and
  falls back to CBC via PuLP — neither is available here, and the north
  star mandates an open solver).
- the time model is xGMI-native: all directed channels transfer
  CONCURRENTLY (point-to-point links), so exchange time = max over
  channels of alpha_ch * MB_ch + beta_ch (the reference models gloo ring
  rounds, ``assigner.py:364-377``).
- aggregation-score computation is a vectorized index_add over the CSR
  plus one reverse all-to-all (the reference all-gathers score maps,
  ``processing.py:40-107``).

MILP per exchange key:
    min  lam * nvar(x) + (1-lam) * ntime(Z)
    s.t. sum_b x[ch,g,b] = 1            (one bit width per group)
         Z >= alpha_ch * sum_g,b bytes(g,b) x[ch,g,b] + beta_ch
    x binary, Z >= 0
with nvar/ntime normalized to [0,1] by utopia/nadir points.
"""
from __future__ import annotations

from typing import Dict, List, Optional, Tuple

import numpy as np
import torch
from torch import Tensor

from ..comm.buffers import BITS_SET, bytes_per_node
from ..comm.communicator import Communicator
from ..helpers import AssignScheme, DistGNNType
from .profile import fit_cost_models

BITS_COST = {b: 1.0 / (2.0 ** b - 1) ** 2 for b in BITS_SET}


class Assigner:
    ctx: Optional['Assigner'] = None

    def __init__(self, engine, scheme: AssignScheme, group_size: int = 500,
                 coe_lambda: float = 0.5, init_bits: int = 8,
                 profile_data_length: int = 12, solver_time_limit: float = 30.0):
        self.engine = engine
        self.scheme = scheme
        self.group_size = max(int(group_size), 1)
        self.coe_lambda = float(coe_lambda)
        self.init_bits = int(init_bits)
        self.profile_data_length = profile_data_length
        self.solver_time_limit = solver_time_limit
        self.cost_models: Dict[Tuple[int, int], Tuple[float, float]] = {}
        self.scores: Dict[str, Tensor] = {}
        Assigner.ctx = self

    # ------------------------------------------------------------------
    def initial_assignment(self):
        self.engine.set_uniform_assignment(self.init_bits)
        if self.scheme == AssignScheme.ADAPTIVE:
            self.cost_models = fit_cost_models(
                Communicator.ctx, num_points=self.profile_data_length)
            self._compute_scores()
            self.engine.is_tracing = True

    def reassign(self):
        if self.scheme == AssignScheme.UNIFORM:
            return
        if self.scheme == AssignScheme.RANDOM:
            self.engine.set_assignment({
                k: {p: torch.tensor(BITS_SET, dtype=torch.int64)[
                        torch.randint(0, len(BITS_SET), (n,))]
                    for p, n in enumerate(self.engine.graph.send_splits) if n}
                for k in self.engine.exchange_keys()})
            return
        self._adaptive_reassign()

    # ------------------------------------------------------------------
    # aggregation-sensitivity scores
    # ------------------------------------------------------------------
    def _edge_score(self, src_scale: Optional[Tensor],
                    dst_scale: Optional[Tensor]) -> Tensor:
        """score[s] = src_scale[s] * sum_{edges s->d} dst_scale[d] over the
        local CSR; slice [I:] gives each REMOTE node's influence on my
        partition."""
        g = self.engine.graph
        N, I = g.num_nodes, g.num_inner
        counts = (g.indptr[1:] - g.indptr[:-1])
        ds = dst_scale if dst_scale is not None else torch.ones(
            I, device=g.indptr.device)
        per_edge = torch.repeat_interleave(ds, counts)
        acc = torch.zeros(N, device=g.indptr.device)
        acc.index_add_(0, g.indices, per_edge)
        if src_scale is not None:
            acc = acc * src_scale
        return acc

    def _compute_scores(self):
        """For every key, a score per (peer, send-node) aligned with the
        total_send_idx layout — obtained by computing each REMOTE node's
        aggregation weight locally and shipping it back to the owner."""
        e = self.engine
        g = e.graph
        comm = Communicator.ctx
        if e.model_type == DistGNNType.DistGCN:
            fwd = self._edge_score(e.gcn_src_f, e.gcn_dst_f)
            bwd = self._edge_score(e.gcn_src_b, e.gcn_dst_b)
        elif e.agg_type == 'mean':
            fwd = self._edge_score(None, e.sage_dst_f)
            bwd = self._edge_score(e.sage_src_b, None)
        else:
            fwd = self._edge_score(None, e.sage1_dst_f)
            bwd = self._edge_score(e.sage1_src_b, None)
        for name, vec in (('forward', fwd), ('backward', bwd)):
            remote_scores = vec[g.num_inner:].contiguous().cpu()
            # reverse exchange: remote-block scores -> owners' send layout
            out = torch.empty(g.num_send, dtype=remote_scores.dtype)
            torch.distributed.all_to_all_single(
                out, remote_scores, list(g.send_splits), list(g.recv_splits))
            for l in range(e.num_layers):
                key = f'{name}{l}'
                if key in e.exchange_keys():
                    self.scores[key] = out.to(e.device)

    # ------------------------------------------------------------------
    # adaptive pipeline
    # ------------------------------------------------------------------
    def _group_key(self, key: str):
        """Per peer: sort nodes by combined variance desc, group, and emit
        (order, counts, var[G,B], mbytes[G,B])."""
        e = self.engine
        g = e.graph
        S = g.num_send
        traced = e.traced.get(key)
        traced = traced if traced is not None else torch.ones(S, device=e.device)
        score = self.scores.get(key, torch.ones(S, device=e.device))
        combined = (score.float() ** 2) * traced.float()
        F = e.key_dim(key)
        out = {}
        off = 0
        for p, n in enumerate(g.send_splits):
            if n == 0:
                continue
            c = combined[off:off + n]
            order = torch.argsort(c, descending=True).cpu()
            csorted = c.cpu()[order]
            G = (n + self.group_size - 1) // self.group_size
            var = np.zeros((G, len(BITS_SET)))
            mb = np.zeros((G, len(BITS_SET)))
            counts = []
            for gi in range(G):
                sl = csorted[gi * self.group_size:(gi + 1) * self.group_size]
                cnt = int(sl.numel())
                counts.append(cnt)
                s = float(sl.sum())
                for bi, b in enumerate(BITS_SET):
                    var[gi, bi] = BITS_COST[b] * s
                    mb[gi, bi] = cnt * (bytes_per_node(F, b) + 4) / 2 ** 20
            out[p] = {'order': order, 'counts': counts, 'var': var, 'mb': mb}
            off += n
        return out

    def _adaptive_reassign(self):
        e = self.engine
        comm = Communicator.ctx
        rank, W = comm.rank, comm.world_size
        keys = e.exchange_keys()
        grouped = {k: self._group_key(k) for k in keys}
        payload = {k: {p: {'counts': v['counts'], 'var': v['var'], 'mb': v['mb']}
                       for p, v in grouped[k].items()} for k in keys}
        gathered = comm.gather_object({'groups': payload,
                                       'cost': self.cost_models}, dst=0)
        if rank == 0:
            cost = {}
            for gobj in gathered:
                cost.update(gobj['cost'])
            solution = {k: self._solve_key(k, [gobj['groups'][k] for gobj in gathered],
                                           cost) for k in keys}
            per_rank = [{k: solution[k][r] for k in keys} for r in range(W)]
        else:
            per_rank = None
        mine = comm.scatter_object(per_rank, src=0)
        assignments = {}
        for k in keys:
            assignments[k] = {}
            for p, info in grouped[k].items():
                gb = mine[k].get(p)
                n = sum(info['counts'])
                bits_vec = torch.empty(n, dtype=torch.int64)
                o = 0
                for gi, cnt in enumerate(info['counts']):
                    b = BITS_SET[gb[gi]] if gb is not None else self.init_bits
                    bits_vec[info['order'][o:o + cnt]] = b
                    o += cnt
                assignments[k][p] = bits_vec
        e.set_assignment(assignments)
        e.reset_trace()

    def _solve_key(self, key: str, rank_groups: List[dict], cost
                   ) -> List[Dict[int, List[int]]]:
        """Solve one key's MILP on rank 0. rank_groups[r] = {peer: {counts,
        var[G,B], mb[G,B]}}. Returns per-rank {peer: [bit_index per group]}."""
        from scipy import sparse
        from scipy.optimize import LinearConstraint, Bounds, milp

        B = len(BITS_SET)
        channels = []   # (rank, peer, var, mb, alpha, beta)
        for r, groups in enumerate(rank_groups):
            for p, info in groups.items():
                a, b = cost.get((r, p), (1e-3, 1e-5))
                channels.append((r, p, info['var'], info['mb'], a, b))
        nx = sum(ch[2].shape[0] * B for ch in channels)
        if nx == 0:
            return [dict() for _ in rank_groups]
        nvar = nx + 1                         # + Z
        var_coef = np.zeros(nx)
        rows_eq, cols_eq = [], []
        A_time_rows = []
        time_rhs = []
        off = 0
        eq = 0
        t_lo = 0.0
        t_hi = 0.0
        for (_, _, var, mb, a, beta) in channels:
            G = var.shape[0]
            var_coef[off:off + G * B] = var.reshape(-1)
            row = np.zeros(nvar)
            for gi in range(G):
                cols = off + gi * B + np.arange(B)
                rows_eq.extend([eq] * B)
                cols_eq.extend(cols.tolist())
                eq += 1
            row[off:off + G * B] = (a * mb).reshape(-1)
            row[-1] = -1.0
            A_time_rows.append(row)
            time_rhs.append(-beta)
            t_hi = max(t_hi, a * mb[:, -1].sum() + beta)   # all 8-bit
            t_lo = max(t_lo, a * mb[:, 0].sum() + beta)    # all 2-bit
            off += G * B
        v_hi = sum(ch[2][:, 0].sum() for ch in channels)   # all 2-bit
        v_lo = sum(ch[2][:, -1].sum() for ch in channels)  # all 8-bit
        v_rng = max(v_hi - v_lo, 1e-12)
        t_rng = max(t_hi - t_lo, 1e-12)

        c = np.zeros(nvar)
        c[:nx] = self.coe_lambda * var_coef / v_rng
        c[-1] = (1.0 - self.coe_lambda) / t_rng

        A_eq = sparse.coo_matrix((np.ones(len(rows_eq)), (rows_eq, cols_eq)),
                                 shape=(eq, nvar))
        con_eq = LinearConstraint(A_eq, 1.0, 1.0)
        A_t = np.vstack(A_time_rows)
        con_t = LinearConstraint(sparse.coo_matrix(A_t), -np.inf,
                                 np.array(time_rhs))
        integrality = np.concatenate([np.ones(nx), np.zeros(1)])
        bounds = Bounds(np.zeros(nvar),
                        np.concatenate([np.ones(nx), [np.inf]]))
        res = milp(c=c, constraints=[con_eq, con_t], integrality=integrality,
                   bounds=bounds, options={'time_limit': self.solver_time_limit})
        out = [dict() for _ in rank_groups]
        if res.x is None:
            # solver failed -> keep uniform init bits
            bi_init = BITS_SET.index(self.init_bits)
            off = 0
            for (r, p, var, mb, _, _) in channels:
                G = var.shape[0]
                out[r][p] = [bi_init] * G
                off += G * B
            return out
        x = res.x[:nx]
        off = 0
        for (r, p, var, mb, _, _) in channels:
            G = var.shape[0]
            choice = x[off:off + G * B].reshape(G, B).argmax(axis=1)
            out[r][p] = choice.tolist()
            off += G * B
        return out
