"""Mixed-bit boundary-exchange layout plans.

Reference parity: ``AdaQP/communicator/buffer.py`` (pinned-CPU + GPU
train/test buffers, per-bit index bookkeeping exchanged by
``all_gather_object``, rebuilt on every re-assignment) and the mixed
quantization glue ``AdaQP/model/op_util.py:189-236``.

MI355X redesign: everything stays on the GPU, so "buffers" reduce to a
LAYOUT PLAN — flat per-bit index/offset tensors that drive ONE fused
quant (or dequant) HIP kernel launch per bit-width and ONE
``all_to_all_single`` per payload. Wire format for each key
(``forward{l}``/``backward{l}``):

    payload uint8, peer-major; within a peer: bit groups in (2,4,8)
    order; within a group: the agreed global-id node order; each node
    occupies ceil(F*bits/8) bytes.
    params  bf16, 2 per node (scale, rmin), same node order.

Both sides derive the same layout from the per-node bit widths, which
the sender distributes with one int8 all-to-all (``exchange_bits``)
after every re-assignment — replacing the reference's
``all_gather_object`` of nested index dicts (``buffer.py:219-231``).
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import Dict, List, Optional

import torch
from torch import Tensor

from ..graph import LocalGraph
from .communicator import Communicator

BITS_SET = (2, 4, 8)


def bytes_per_node(F: int, bits: int) -> int:
    return (F * bits + 7) // 8


@dataclass
class SidePlan:
    """One direction (send or recv) of a key's wire layout."""
    F: int
    rows: Dict[int, Tensor]       # bit -> int64 row ids (local x rows / remote-block rows)
    pos: Dict[int, Tensor]        # bit -> int64 wire node positions
    off: Dict[int, Tensor]        # bit -> int64 wire byte offsets
    byte_splits: List[int]        # per peer payload bytes
    node_splits: List[int]        # per peer node counts
    total_bytes: int
    total_nodes: int

    @property
    def param_splits(self) -> List[int]:
        return [2 * n for n in self.node_splits]

    def to(self, device) -> 'SidePlan':
        self.rows = {b: t.to(device) for b, t in self.rows.items()}
        self.pos = {b: t.to(device) for b, t in self.pos.items()}
        self.off = {b: t.to(device) for b, t in self.off.items()}
        return self


@dataclass
class KeyPlan:
    """Quantized-exchange plan for one (layer, direction) key."""
    F: int
    send: SidePlan
    recv: SidePlan

    def to(self, device) -> 'KeyPlan':
        self.send.to(device)
        self.recv.to(device)
        return self


def _layout(bits_per_peer: List[Optional[Tensor]],
            rows_per_peer: List[Optional[Tensor]], F: int) -> SidePlan:
    """Build one side's layout from per-peer bit vectors (agreed order)."""
    rows = {b: [] for b in BITS_SET}
    pos = {b: [] for b in BITS_SET}
    off = {b: [] for b in BITS_SET}
    byte_splits, node_splits = [], []
    p_cursor = 0
    b_cursor = 0
    for bits_vec, rows_vec in zip(bits_per_peer, rows_per_peer):
        n = 0 if bits_vec is None else int(bits_vec.numel())
        if n == 0:
            byte_splits.append(0)
            node_splits.append(0)
            continue
        peer_bytes = 0
        for b in BITS_SET:
            sel = torch.nonzero(bits_vec == b, as_tuple=True)[0]
            k = int(sel.numel())
            if k == 0:
                continue
            bpn = bytes_per_node(F, b)
            rows[b].append(rows_vec[sel])
            pos[b].append(torch.arange(k, dtype=torch.int64) + p_cursor)
            off[b].append(torch.arange(k, dtype=torch.int64) * bpn + b_cursor)
            p_cursor += k
            b_cursor += k * bpn
            peer_bytes += k * bpn
        if p_cursor - sum(node_splits) != n:
            bad = set(torch.unique(bits_vec).tolist()) - set(BITS_SET)
            raise ValueError(f'invalid bit widths in assignment: {bad}')
        byte_splits.append(peer_bytes)
        node_splits.append(n)
    cat = lambda d: {b: (torch.cat(v) if v else torch.empty(0, dtype=torch.int64))
                     for b, v in d.items()}
    return SidePlan(F, cat(rows), cat(pos), cat(off), byte_splits, node_splits,
                    b_cursor, p_cursor)


def exchange_bits(comm: Communicator, graph: LocalGraph,
                  send_bits: Dict[int, Tensor]) -> List[Optional[Tensor]]:
    """Distribute my per-node bit widths to consumers; returns per-peer bit
    vectors for my remote block (CPU int8 gloo all-to-all)."""
    send_parts = []
    for p in range(graph.world_size):
        if graph.send_splits[p]:
            send_parts.append(send_bits[p].to(torch.uint8).cpu())
    send = torch.cat(send_parts) if send_parts else torch.empty(0, dtype=torch.uint8)
    recv = torch.empty(sum(graph.recv_splits), dtype=torch.uint8)
    torch.distributed.all_to_all_single(
        recv, send, list(graph.recv_splits), list(graph.send_splits))
    out: List[Optional[Tensor]] = []
    o = 0
    for p in range(graph.world_size):
        n = graph.recv_splits[p]
        out.append(recv[o:o + n].to(torch.int64) if n else None)
        o += n
    return out


def build_key_plan(comm: Communicator, graph: LocalGraph, F: int,
                   send_bits: Dict[int, Tensor],
                   device=None) -> KeyPlan:
    """send_bits: peer -> int64 bit widths over send_idx[peer] order."""
    send_bits_l = [send_bits.get(p) if p != graph.rank and graph.send_splits[p]
                   else None for p in range(graph.world_size)]
    send_rows_l = [graph.send_idx[p].cpu() if graph.send_splits[p] else None
                   for p in range(graph.world_size)]
    recv_bits_l = exchange_bits(comm, graph, send_bits)
    base = 0
    recv_rows_l: List[Optional[Tensor]] = []
    for p in range(graph.world_size):
        n = graph.recv_splits[p]
        recv_rows_l.append(torch.arange(base, base + n, dtype=torch.int64) if n else None)
        base += n
    plan = KeyPlan(F, _layout(send_bits_l, send_rows_l, F),
                   _layout(recv_bits_l, recv_rows_l, F))
    if device is not None:
        plan.to(device)
    return plan


def uniform_bits(graph: LocalGraph, bits: int) -> Dict[int, Tensor]:
    return {p: torch.full((graph.send_splits[p],), bits, dtype=torch.int64)
            for p in range(graph.world_size) if graph.send_splits[p]}
