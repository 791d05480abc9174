"""Distributed aggregation autograd ops.

Reference parity: ``AdaQP/model/ops.py`` (DistAggConv / DistAggSAGE,
full_graph_propagation / decomposed_graph_propagation) and the exchange
glue ``AdaQP/model/op_util.py:138-236``.

MI355X redesign of the overlap (reference: side CPU thread + 2 CUDA
events + 2 CPU events, ``ops.py:119-130``, ``op_util.py:101-130``):
RCCL collectives are stream-ordered device work, so the whole
quantize -> all_to_all -> dequantize pipeline is enqueued on a side HIP
stream while the central aggregation runs on the default stream; one
cuda event orders dequant-output -> marginal aggregation. No threads.

Backward uses the SAME exchange direction and the SAME CSR as forward
(bidirected graphs): the grad of my inner boundary nodes is sent out,
remote grads are received, and the in-edge aggregation with swapped
normalization performs the cross-partition reduction — exact transpose
(see ops.py:17-32 for the reference's equivalent on dgl.reverse graphs).
"""
from __future__ import annotations

from typing import Optional, Tuple

import torch
from torch import Tensor

from ..helpers import BitType, DistGNNType, PropagationMode
from ..comm.communicator import Communicator
from .kernels import mixed_quantize, mixed_dequantize, spmm


# --------------------------------------------------------------------------
# boundary exchange
# --------------------------------------------------------------------------

def _exchange_start(engine, x_local: Tensor, key: str, quant: bool):
    """Device-side producer work only (gather or quantize kernels) — no
    transport. Split from :func:`_exchange_finish` so the decomposed path
    can enqueue the central aggregation BETWEEN the two: the transport may
    block the host (gloo staging), and anything enqueued after it would
    lose the overlap."""
    if quant:
        plan = engine.plans[key]
        seed = engine.next_seed()
        if engine.is_tracing:
            send = x_local.index_select(0, engine.graph.total_send_idx)
            engine.trace(key, send)
        with engine.timer.record(f'{key}_quant'):
            payload, params = mixed_quantize(x_local, plan.send, seed)
        return ('qt', payload, params, x_local.dtype)
    send = x_local.index_select(0, engine.graph.total_send_idx)
    if engine.is_tracing:
        engine.trace(key, send)
    return ('fp', send, None, x_local.dtype)


def _exchange_finish(engine, staged, key: str) -> Tensor:
    """Transport + consumer-side kernels (dequant/scatter)."""
    comm = Communicator.ctx
    g = engine.graph
    kind, a, b, dtype = staged
    if kind == 'fp':
        with engine.timer.record(f'{key}_exchange'):
            recv, _ = comm.exchange_rows(a, g.send_splits, g.recv_splits)
        return recv
    plan = engine.plans[key]
    payload, params = a, b
    recv_payload = torch.empty(plan.recv.total_bytes, dtype=torch.uint8,
                               device=payload.device)
    recv_params = torch.empty(2 * plan.recv.total_nodes, dtype=torch.bfloat16,
                              device=payload.device)
    with engine.timer.record(f'{key}_exchange'):
        comm.all_to_all_v(recv_payload, payload,
                          plan.recv.byte_splits, plan.send.byte_splits)
        comm.all_to_all_v(recv_params, params,
                          plan.recv.param_splits, plan.send.param_splits)
    out = torch.empty(g.num_remote, plan.F, dtype=dtype,
                      device=payload.device)
    with engine.timer.record(f'{key}_dequant'):
        mixed_dequantize(recv_payload, recv_params, plan.recv, out)
    return out


def fp_exchange(engine, x_local: Tensor, key: str) -> Tensor:
    return _exchange_finish(engine,
                            _exchange_start(engine, x_local, key, False), key)


def qt_exchange(engine, x_local: Tensor, key: str) -> Tensor:
    return _exchange_finish(engine,
                            _exchange_start(engine, x_local, key, True), key)


def _exchange(engine, x_local: Tensor, key: str, is_train: bool) -> Tensor:
    if engine.bit_type == BitType.QUANT and is_train:
        return qt_exchange(engine, x_local, key)
    return fp_exchange(engine, x_local, key)


# --------------------------------------------------------------------------
# aggregation (SpMM with model-specific normalization)
# --------------------------------------------------------------------------

def _scales(engine, mode: PropagationMode) -> Tuple[Optional[Tensor], Optional[Tensor], bool]:
    """(src_scale[N], dst_scale[I], add_self) for the model/mode."""
    fwd = mode == PropagationMode.Forward
    if engine.model_type == DistGNNType.DistGCN:
        return ((engine.gcn_src_f, engine.gcn_dst_f, False) if fwd
                else (engine.gcn_src_b, engine.gcn_dst_b, False))
    if engine.agg_type == 'mean':
        return ((None, engine.sage_dst_f, False) if fwd
                else (engine.sage_src_b, None, False))
    if engine.agg_type == 'gcn':
        return ((None, engine.sage1_dst_f, True) if fwd
                else (engine.sage1_src_b, None, True))
    raise ValueError(f'unknown aggregator {engine.agg_type}')


def _agg(engine, view, x_local: Tensor, x_remote, src_scale, dst_scale,
         out=None) -> Tensor:
    """SpMM over one SpmmView. src_scale is the full [N] vector; dst_scale
    is sliced to the view's row range."""
    ds = (dst_scale[view.base:view.base + view.nrows]
          if dst_scale is not None else None)
    return spmm(view, x_local, x_remote, src_scale, ds, out=out)


def _self_term(engine, x_local: Tensor, mode: PropagationMode) -> Tensor:
    """SAGE 'gcn' aggregator self term (reference ops.py:44-47,57-64)."""
    I = engine.graph.num_inner
    scale = (engine.sage1_dst_f if mode == PropagationMode.Forward
             else engine.sage1_src_b[:I])
    return (x_local[:I] * scale[:, None]).to(x_local.dtype)


# --------------------------------------------------------------------------
# propagation paths
# --------------------------------------------------------------------------

def full_propagation(engine, x_local: Tensor, key: str, is_train: bool,
                     mode: PropagationMode) -> Tensor:
    remote = _exchange(engine, x_local, key, is_train)
    src_scale, dst_scale, add_self = _scales(engine, mode)
    with engine.timer.record(f'{key}_full_aggregation'):
        y = _agg(engine, engine.full_view, x_local, remote, src_scale, dst_scale)
        if add_self:
            y = y + _self_term(engine, x_local, mode)
    return y


def decomposed_propagation(engine, x_local: Tensor, key: str, is_train: bool,
                           mode: PropagationMode) -> Tensor:
    """Central aggregation (default stream) overlapped with
    quant->all_to_all->dequant (comm stream).

    Host-call order matters: producer kernels (quantize/gather) are
    enqueued on the comm stream, then the central SpMM is enqueued on the
    default stream, and only THEN the transport runs. With RCCL the
    transport is stream-ordered device work so any order would overlap;
    with a host-blocking transport (gloo staging — several ranks sharing
    one GPU, or CPU debug) the central SpMM must already be in the GPU
    queue when the host blocks, otherwise the overlap the decomposition
    exists for never happens (the reference needed a side thread for
    this, ``ops.py:156-193``; stream order + one event replaces it)."""
    src_scale, dst_scale, add_self = _scales(engine, mode)
    g = engine.graph
    on_gpu = x_local.is_cuda
    quant = engine.bit_type == BitType.QUANT and is_train

    if on_gpu:
        cur = torch.cuda.current_stream()
        engine.comm_stream.wait_stream(cur)          # x_local is ready
        x_local.record_stream(engine.comm_stream)
        with torch.cuda.stream(engine.comm_stream):
            staged = _exchange_start(engine, x_local, key, quant)
        # central and marginal SpMMs write disjoint row slices of ONE
        # output tensor (no torch.cat: the cat was a full [I,F]
        # read+write per propagation)
        C = g.num_central
        y = torch.empty(g.num_inner, x_local.shape[1], dtype=x_local.dtype,
                        device=x_local.device)
        # central rows only touch local columns -> enqueue NOW, before
        # the transport can block the host
        with engine.timer.record(f'{key}_central_aggregation'):
            _agg(engine, engine.central_view, x_local, None, src_scale,
                 dst_scale, out=y[:C])
        with torch.cuda.stream(engine.comm_stream):
            remote = _exchange_finish(engine, staged, key)
            engine.remote_ready.record(engine.comm_stream)
        torch.cuda.current_stream().wait_event(engine.remote_ready)
        remote.record_stream(torch.cuda.current_stream())
        with engine.timer.record(f'{key}_marginal_aggregation'):
            _agg(engine, engine.marginal_view, x_local, remote, src_scale,
                 dst_scale, out=y[C:])
    else:
        remote = _exchange(engine, x_local, key, is_train)
        with engine.timer.record(f'{key}_central_aggregation'):
            y_c = _agg(engine, engine.central_view, x_local, None, src_scale,
                       dst_scale)
        with engine.timer.record(f'{key}_marginal_aggregation'):
            y_m = _agg(engine, engine.marginal_view, x_local, remote,
                       src_scale, dst_scale)
        y = torch.cat([y_c, y_m], dim=0)
    if add_self:
        y = y + _self_term(engine, x_local, mode)
    return y


def propagate(engine, x_local: Tensor, layer: int, is_train: bool,
              mode: PropagationMode) -> Tensor:
    key = (f'forward{layer}' if mode == PropagationMode.Forward
           else f'backward{layer}')
    if engine.use_parallel:
        return decomposed_propagation(engine, x_local, key, is_train, mode)
    return full_propagation(engine, x_local, key, is_train, mode)


# --------------------------------------------------------------------------
# autograd Function (one class serves GCN and SAGE; the engine carries the
# model type — reference has DistAggConv/DistAggSAGE, ops.py:69-111)
# --------------------------------------------------------------------------

class DistAgg(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x_local: Tensor, engine, layer: int, is_train: bool) -> Tensor:
        ctx.engine = engine
        ctx.layer = layer
        ctx.in_dtype = x_local.dtype
        x_local = x_local.to(engine.compute_dtype).contiguous()
        return propagate(engine, x_local, layer, is_train, PropagationMode.Forward)

    @staticmethod
    def backward(ctx, grad_out: Tensor):
        g = grad_out.to(ctx.engine.compute_dtype).contiguous()
        gx = propagate(ctx.engine, g, ctx.layer, True, PropagationMode.Backward)
        return gx.to(ctx.in_dtype), None, None, None


def dist_aggregate(x_local: Tensor, engine, layer: int, is_train: bool) -> Tensor:
    return DistAgg.apply(x_local, engine, layer, is_train)
