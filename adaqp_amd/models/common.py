"""FastLinear: x @ W + b with a GEMM-based bias gradient.

torch's autograd computes grad_bias as ``grad.sum(dim=0)``, which on
ROCm lowers to a strided column reduction running at ~0.2 TB/s for the
[N, 256] grads this model produces (measured: 3.8 ms for 0.73M x 256 —
~10% of an epoch). Computing it as ``ones[1,N] @ grad`` instead routes
it through rocBLAS at memory bandwidth.
"""
from __future__ import annotations

import torch
import torch.nn as nn
from torch import Tensor

_ones_cache: dict = {}


def _ones_row(n: int, device, dtype) -> Tensor:
    key = (n, device, dtype)
    t = _ones_cache.get(key)
    if t is None or t.shape[1] < n:
        t = torch.ones(1, n, device=device, dtype=dtype)
        if len(_ones_cache) > 16:
            _ones_cache.clear()
        _ones_cache[key] = t
    return t[:, :n]


class _LinearBiasFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x: Tensor, w: Tensor, b: Tensor) -> Tensor:
        ctx.save_for_backward(x, w)
        return torch.addmm(b, x, w)

    @staticmethod
    def backward(ctx, g: Tensor):
        x, w = ctx.saved_tensors
        g = g.contiguous()
        gx = g @ w.t() if ctx.needs_input_grad[0] else None
        gw = x.t() @ g if ctx.needs_input_grad[1] else None
        gb = None
        if ctx.needs_input_grad[2]:
            if g.is_cuda:
                gb = (_ones_row(g.shape[0], g.device, g.dtype) @ g).reshape(-1)
            else:
                gb = g.sum(dim=0)
        return gx, gw, gb


class FastLinear(nn.Module):
    """Drop-in linear (weight stored [in, out]) with fast bias grad."""

    def __init__(self, in_dim: int, out_dim: int, bias: bool = True):
        super().__init__()
        self.weight = nn.Parameter(torch.empty(in_dim, out_dim))
        self.bias = nn.Parameter(torch.zeros(out_dim)) if bias else None
        self.reset_parameters()

    def reset_parameters(self):
        nn.init.xavier_uniform_(self.weight)
        if self.bias is not None:
            nn.init.zeros_(self.bias)

    def forward(self, x: Tensor) -> Tensor:
        w, b = self.weight, self.bias
        if torch.is_autocast_enabled('cuda') and x.is_cuda:
            # differentiable casts OUTSIDE the Function: bf16 compute,
            # fp32 master weights get fp32 grads via the cast backward
            dt = torch.get_autocast_dtype('cuda')
            x = x.to(dt)
            w = w.to(dt)
            b = b.to(dt) if b is not None else None
        if b is None:
            return x @ w
        return _LinearBiasFn.apply(x, w, b)


class _FusedDualLinearFn(torch.autograd.Function):
    """out = x @ w1 + h @ w2 + (b1 + b2) via the one-pass MFMA kernel
    (csrc fused_dual_gemm_bf16); backward through plain GEMMs."""

    @staticmethod
    def forward(ctx, x: Tensor, h: Tensor, w1: Tensor, w2: Tensor,
                b1: Tensor, b2: Tensor) -> Tensor:
        from ..ops.kernels import native
        ctx.save_for_backward(x, h, w1, w2)
        M, N = x.shape[0], w1.shape[1]
        out = torch.empty(M, N, dtype=x.dtype, device=x.device)
        bias = ((b1 + b2).to(x.dtype) if b1 is not None
                else torch.empty(0, dtype=x.dtype, device=x.device))
        native().fused_dual_gemm_bf16(x.contiguous(), h.contiguous(),
                                      w1.t().contiguous(), w2.t().contiguous(),
                                      bias, out)
        return out

    @staticmethod
    def backward(ctx, g: Tensor):
        x, h, w1, w2 = ctx.saved_tensors
        g = g.contiguous()
        gx = g @ w1.t() if ctx.needs_input_grad[0] else None
        gh = g @ w2.t() if ctx.needs_input_grad[1] else None
        gw1 = x.t() @ g if ctx.needs_input_grad[2] else None
        gw2 = h.t() @ g if ctx.needs_input_grad[3] else None
        gb = None
        if ctx.needs_input_grad[4]:
            gb = (_ones_row(g.shape[0], g.device, g.dtype) @ g).reshape(-1)
        return gx, gh, gw1, gw2, gb, gb


def fused_dual_linear_ok(x: Tensor, h: Tensor, n_out: int) -> bool:
    """Opt-in (ADAQP_FUSED_SAGE=1): measured SLOWER than two TUNED
    hipBLASLt GEMMs + add on the products shape (152-177 vs 132 ms/epoch
    — profiles/r01_NOTES.md); hipBLASLt's pipelined schedules win this
    bandwidth-bound tall-skinny shape. Kept as a tested reference
    implementation of a fused MFMA epilogue."""
    import os
    if os.environ.get('ADAQP_FUSED_SAGE') != '1':
        return False
    from ..ops.kernels import has_native
    # fused_dual_linear casts both operands to bf16, so fp32 inputs are
    # acceptable (layer-0 features stay fp32 under a bf16 engine).
    ok_dt = (torch.bfloat16, torch.float32)
    return (x.is_cuda and x.dtype in ok_dt and h.dtype in ok_dt
            and n_out % 16 == 0 and n_out <= 256
            and x.shape[1] % 8 == 0 and h.shape[1] % 8 == 0
            and has_native())


def fused_dual_linear(x, h, lin1: 'FastLinear', lin2: 'FastLinear') -> Tensor:
    """lin1(x) + lin2(h) in one MFMA pass (bf16 CUDA); casts mirror
    FastLinear.forward so fp32 master weights get fp32 grads."""
    dt = torch.bfloat16
    return _FusedDualLinearFn.apply(
        x.to(dt), h.to(dt), lin1.weight.to(dt), lin2.weight.to(dt),
        lin1.bias.to(dt) if lin1.bias is not None else None,
        lin2.bias.to(dt) if lin2.bias is not None else None)
