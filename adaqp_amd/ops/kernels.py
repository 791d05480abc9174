"""Device dispatch for the hot ops.

GPU (MI355X): hand-written CDNA4 HIP kernels from ``adaqp_amd/csrc``
(built in-tree as ``adaqp_amd._C``). The GPU path REFUSES to fall back —
if the extension is missing on a CUDA device, that is a build error, not
a reason to silently run eager PyTorch.

CPU: torch reference implementations (bit-exact for quant pack/unpack —
same hash RNG) used by unit tests and the CPU/gloo plumbing config.
"""
from __future__ import annotations

from typing import Optional, Tuple

import torch
from torch import Tensor

from . import quant as Q
from ..comm.buffers import SidePlan, BITS_SET, bytes_per_node

_native = None
_native_err: Optional[Exception] = None


def native():
    global _native, _native_err
    if _native is None and _native_err is None:
        try:
            from .. import _C  # in-tree HIP extension
            _native = _C
        except Exception as e:   # pragma: no cover
            _native_err = e
    if _native is None:
        raise RuntimeError(
            f'adaqp_amd._C HIP extension not built (run `python setup.py '
            f'build_ext --inplace`): {_native_err}')
    return _native


def has_native() -> bool:
    try:
        native()
        return True
    except RuntimeError:
        return False


# --------------------------------------------------------------------------
# mixed-bit quantize / dequantize
# --------------------------------------------------------------------------

def mixed_quantize(x: Tensor, plan: SidePlan, seed: int) -> Tuple[Tensor, Tensor]:
    """Gather plan.rows from x, quantize per bit group, pack into the wire
    payload + params described by plan. Returns (uint8 [B], bf16 [2S])."""
    if x.is_cuda:
        payload = torch.empty(plan.total_bytes, dtype=torch.uint8, device=x.device)
        params = torch.empty(2 * plan.total_nodes, dtype=torch.bfloat16, device=x.device)
        for b in BITS_SET:
            rows = plan.rows[b]
            if rows.numel():
                native().quant_pack(x, rows, plan.pos[b], plan.off[b], b,
                                    seed, payload, params)
        return payload, params
    payload = torch.zeros(plan.total_bytes, dtype=torch.uint8)
    params = torch.zeros(2 * plan.total_nodes, dtype=torch.bfloat16)
    for b in BITS_SET:
        rows, pos, off = plan.rows[b], plan.pos[b], plan.off[b]
        if rows.numel() == 0:
            continue
        pl, scale, rmin = Q.pack_torch(x[rows].float(), b, seed, node_tag=pos)
        bpn = pl.shape[1]
        idx = off[:, None] + torch.arange(bpn, dtype=torch.int64)
        payload[idx.reshape(-1)] = pl.reshape(-1)
        params[2 * pos] = scale
        params[2 * pos + 1] = rmin
    return payload, params


def mixed_dequantize(payload: Tensor, params: Tensor, plan: SidePlan,
                     out: Tensor) -> Tensor:
    """Unpack wire payload into fp rows of ``out`` (remote block [R, F]),
    fused with the scatter to plan.rows."""
    if payload.is_cuda:
        for b in BITS_SET:
            rows = plan.rows[b]
            if rows.numel():
                native().quant_unpack(payload, params, rows, plan.pos[b],
                                      plan.off[b], b, plan.F, out)
        return out
    for b in BITS_SET:
        rows, pos, off = plan.rows[b], plan.pos[b], plan.off[b]
        if rows.numel() == 0:
            continue
        bpn = bytes_per_node(plan.F, b)
        idx = off[:, None] + torch.arange(bpn, dtype=torch.int64)
        pl = payload[idx.reshape(-1)].reshape(-1, bpn)
        scale = params[2 * pos].clone()
        rmin = params[2 * pos + 1].clone()
        out[rows] = Q.unpack_torch(pl, b, scale, rmin, plan.F).to(out.dtype)
    return out


# --------------------------------------------------------------------------
# CSR SpMM:  y[r] = dst_scale[r] * sum_{c in row r} src_scale[c] * x[c]
# --------------------------------------------------------------------------

_csr_cache: dict = {}

import os as _os
# max edges one sub-wavefront processes serially. 256 measured best on
# the flagship shape in a same-box sweep (128/192/256 flat, 512/1024
# slower); cross-box deltas on other datasets are within run-to-run
# noise (profiles/r02_NOTES.md). ADAQP_SEG_EDGES overrides.
SEG_EDGES = int(_os.environ.get('ADAQP_SEG_EDGES', '256'))


def _auto_seg(num_edges: int, num_rows: int) -> int:
    return SEG_EDGES


class SpmmView:
    """One CSR row range + its work-item segmentation.

    Power-law graphs put 10^4-10^5 edges on hub rows; a single
    sub-wavefront grinding one such row serially becomes the kernel's
    critical path. Rows are split into <=SEG_EDGES-edge segments; rows
    with >1 segment combine via global atomics into pre-zeroed output
    rows (SURVEY.md §7 'hard parts': row-binning / load imbalance).

    ``col_block`` > 0 additionally PERMUTES the edge array into
    column-block-major order (segments keyed by (col/col_block, row)):
    the grid sweeps one x column block at a time, so the block stays
    LLC-resident while every consumer reads it — same kernel, different
    work-item layout. Costs atomic combines for every row spanning >1
    block.
    """

    def __init__(self, indptr: Tensor, indices: Tensor, base: int, nrows: int,
                 col_block: int = 0):
        self.indptr = indptr
        self.base = int(base)
        self.nrows = int(nrows)
        self.col_blocked = col_block > 0 and indices.numel() > 0
        SEG = _auto_seg(int(indices.numel()), nrows)
        indptr_c = indptr.cpu()
        counts = (indptr_c[1:] - indptr_c[:-1])
        if col_block > 0 and indices.numel():
            idx_c = indices.cpu()
            rows_pe = torch.repeat_interleave(
                torch.arange(nrows, dtype=torch.int64), counts)
            cb = idx_c // col_block
            key = cb * nrows + rows_pe
            order = torch.argsort(key, stable=True)
            self.indices = idx_c[order].to(indices.device)
            key = key[order]
            # run starts: key change points, then chunk runs to SEG_EDGES
            change = torch.ones(key.numel(), dtype=torch.bool)
            change[1:] = key[1:] != key[:-1]
            run_start = torch.nonzero(change, as_tuple=True)[0]
            run_end = torch.cat([run_start[1:],
                                 torch.tensor([key.numel()])])
            run_len = run_end - run_start
            nseg = (run_len + SEG - 1) // SEG
            seg_runs = torch.repeat_interleave(
                torch.arange(run_start.numel()), nseg)
            first = torch.repeat_interleave(torch.cumsum(nseg, 0) - nseg, nseg)
            in_run = torch.arange(seg_runs.numel(), dtype=torch.int64) - first
            e0 = run_start[seg_runs] + in_run * SEG
            e1 = torch.minimum(e0 + SEG, run_end[seg_runs])
            seg_row = (key[run_start[seg_runs]] % nrows)
            segs_per_row = torch.bincount(seg_row, minlength=nrows)
            multi_mask = segs_per_row != 1     # incl. empty rows -> zeroed
            self.seg_row = seg_row.to(torch.int32)
            self.seg_e0 = e0
            self.seg_e1 = e1
            self.seg_multi = multi_mask[seg_row].to(torch.uint8)
            self.zero_rows = torch.nonzero(multi_mask, as_tuple=True)[0].to(torch.int32)
        else:
            self.indices = indices
            nseg = torch.clamp((counts + SEG - 1) // SEG, min=1)
            seg_row = torch.repeat_interleave(torch.arange(nrows, dtype=torch.int64), nseg)
            first = torch.repeat_interleave(torch.cumsum(nseg, 0) - nseg, nseg)
            seg_in_row = torch.arange(seg_row.numel(), dtype=torch.int64) - first
            e0 = indptr_c[seg_row] + seg_in_row * SEG
            e1 = torch.minimum(e0 + SEG, indptr_c[seg_row + 1])
            multi_mask = nseg > 1
            self.seg_row = seg_row.to(torch.int32)
            self.seg_e0 = e0
            self.seg_e1 = e1
            self.seg_multi = multi_mask[seg_row].to(torch.uint8)
            self.zero_rows = torch.nonzero(multi_mask, as_tuple=True)[0].to(torch.int32)
        # int32 column indices + segment bounds: halves the kernel's
        # index-read bytes (~1 GB -> 0.5 GB per pass on full products);
        # node/edge counts are always < 2^31
        self.indices = self.indices.to(torch.int32)
        self.seg_e0 = self.seg_e0.to(torch.int32)
        self.seg_e1 = self.seg_e1.to(torch.int32)
        self._on = None

    def to(self, device):
        for n in ('indices', 'seg_row', 'seg_e0', 'seg_e1', 'seg_multi', 'zero_rows'):
            setattr(self, n, getattr(self, n).to(device))
        return self


def spmm(view: 'SpmmView', x_local: Tensor, x_remote: Optional[Tensor],
         src_scale: Optional[Tensor], dst_scale: Optional[Tensor],
         out: Optional[Tensor] = None) -> Tensor:
    """Aggregation SpMM over an in-edge CSR view.

    Columns < len(x_local) read x_local; the rest read x_remote (the
    all-to-all output block) — no concat on the GPU path.
    src_scale: [n_local+n_remote] or None; dst_scale: [nrows] or None.
    ``out`` (GPU path): write into this [nrows, F] contiguous tensor —
    lets the decomposed path target row slices of ONE output so the
    central/marginal results need no torch.cat afterwards.
    """
    indptr, indices, num_rows = view.indptr, view.indices, view.nrows
    if not x_local.is_cuda and view.col_blocked:
        raise RuntimeError('column-blocked SpmmView is GPU-only (the CPU '
                           'fallback consumes indptr, which the blocked '
                           'layout does not preserve)')
    if x_local.is_cuda:
        if out is None:
            out = torch.empty(num_rows, x_local.shape[1], dtype=x_local.dtype,
                              device=x_local.device)
        else:
            assert out.is_contiguous() and out.shape[0] == num_rows
        empty = torch.empty(0, device=x_local.device)
        native().spmm_csr(indices, x_local,
                          x_remote if x_remote is not None else empty, out,
                          src_scale if src_scale is not None else empty,
                          dst_scale if dst_scale is not None else empty,
                          view.seg_row, view.seg_e0, view.seg_e1,
                          view.seg_multi, view.zero_rows)
        return out
    x = (torch.cat([x_local, x_remote], dim=0)
         if x_remote is not None and x_remote.numel() else x_local)
    xs = x * src_scale[:x.shape[0], None] if src_scale is not None else x
    key = (indptr.data_ptr(), indices.data_ptr(), num_rows, x.shape[0])
    sp = _csr_cache.get(key)
    if sp is None:
        crow = indptr.to(torch.int64)
        col = indices.to(torch.int64)
        val = torch.ones(col.numel(), dtype=torch.float32)
        sp = torch.sparse_csr_tensor(crow, col, val, size=(num_rows, x.shape[0]))
        if len(_csr_cache) > 64:
            _csr_cache.clear()
        _csr_cache[key] = sp
    y = torch.sparse.mm(sp, xs.float()).to(x.dtype)
    if dst_scale is not None:
        y = y * dst_scale[:, None]
    if out is not None:
        out.copy_(y)
        return out
    return y
