"""Stochastic integer quantization: torch reference implementation.

Reference parity: ``quant_cuda.pack/unpack_single_precision``
(``/root/reference/AdaQP/util/quantization/src/quantization_cuda_kernel.cu:34-156``)
plus the Python min/max + scale glue (``AdaQP/model/op_util.py:20-67``).

MI355X redesign (the HIP kernels in ``csrc/kernels.hip`` implements the
same math bit-for-bit; this torch version is the CPU path AND the test
oracle):

- ONE fused pass: per-node min/max reduction + scale + stochastic round
  + bit-pack (the reference computes rmin/rmax via two torch reductions
  in Python, then launches the pack kernel).
- packing along the FEATURE axis — node rows stay contiguous byte runs
  (coalesced 128B stores on CDNA4), vs the reference's node-axis packing.
  Each node occupies ceil(F*bits/8) bytes.
- RNG: stateless counter hash of (seed, node_tag, feature) — no curand
  state; the SAME noise sequence on CPU/HIP, so packings are statistically
  identical and byte-identical except where FP contraction (FMA) on the
  GPU lands a value exactly on a rounding boundary (< 0.5% of bytes in
  tests/test_kernels_gpu.py). Harmless: only the sender's kernel ever
  produces a given payload, so there is no cross-device mismatch on the
  wire.
- scale is rounded to bf16 BEFORE quantizing, so the receiver's bf16
  dequant is the exact inverse (the reference quantizes with the fp32
  scale but ships bf16 — a small bias it tolerates).
- values are clamped at BOTH ends (the reference clamps only at 0 and
  relies on the noise term never overflowing — SURVEY.md §2.5).
"""
from __future__ import annotations

from typing import Tuple

import torch
from torch import Tensor

_M32 = 0xFFFFFFFF


def _hash_u32(h: Tensor) -> Tensor:
    """triple32-style avalanche hash on int64 tensors masked to u32."""
    h = h & _M32
    h = (h ^ (h >> 16)) & _M32
    h = (h * 0x7FEB352D) & _M32
    h = (h ^ (h >> 15)) & _M32
    h = (h * 0x846CA68B) & _M32
    h = (h ^ (h >> 16)) & _M32
    return h


def uniform_noise(seed: int, node_tag: Tensor, num_feats: int,
                  device=None) -> Tensor:
    """U[0,1) noise [N, F] from hash(seed, node_tag, feat). node_tag int64 [N]."""
    feat = torch.arange(num_feats, dtype=torch.int64, device=device)
    h = (seed & _M32) ^ ((node_tag[:, None] * 0x9E3779B9) & _M32) \
        ^ ((feat[None, :] * 0x85EBCA6B) & _M32)
    return _hash_u32(h).to(torch.float64).mul_(2.0 ** -32).to(torch.float32)


def qparams(x: Tensor, bits: int) -> Tuple[Tensor, Tensor]:
    """Per-node (scale, rmin), both rounded to bf16 (the wire format).
    scale = (2^b - 1)/(rmax - rmin), or 0 for constant rows."""
    rmin = x.min(dim=1).values
    rmax = x.max(dim=1).values
    rng = rmax - rmin
    scale = torch.where(rng > 0, (2.0 ** bits - 1) / rng.clamp(min=1e-30),
                        torch.zeros_like(rng))
    return scale.to(torch.bfloat16), rmin.to(torch.bfloat16)


def bytes_per_node(num_feats: int, bits: int) -> int:
    """Payload bytes per node; must agree with comm.buffers and the HIP
    kernels' per-node padding (buffer.py:181-186 analogue)."""
    return (num_feats * bits + 7) // 8


def pack_torch(x: Tensor, bits: int, seed: int, node_tag: Tensor
               ) -> Tuple[Tensor, Tensor, Tensor]:
    """Quantize+pack rows of x [N,F] at ``bits`` in {2,4,8}.

    Returns (payload uint8 [N, bytes_per_node], scale bf16 [N], rmin bf16 [N]).
    """
    assert bits in (2, 4, 8)
    N, F = x.shape
    scale, rmin = qparams(x, bits)
    s32 = scale.float()[:, None]
    v = (x - rmin.float()[:, None]) * s32
    u = uniform_noise(seed, node_tag, F, device=x.device)
    q = torch.floor(v + u).clamp_(0, 2 ** bits - 1).to(torch.int64)
    q = torch.where(s32 > 0, q, torch.zeros_like(q))
    vpb = 8 // bits
    Fp = (F + vpb - 1) // vpb * vpb
    if Fp != F:
        q = torch.nn.functional.pad(q, (0, Fp - F))
    q = q.view(N, Fp // vpb, vpb)
    shifts = (torch.arange(vpb, device=x.device, dtype=torch.int64) * bits)
    payload = (q << shifts).sum(dim=2).to(torch.uint8)
    return payload, scale, rmin


def unpack_torch(payload: Tensor, bits: int, scale: Tensor, rmin: Tensor,
                 num_feats: int) -> Tensor:
    """Inverse of pack_torch. payload uint8 [N, bytes_per_node] -> fp32 [N,F]."""
    assert bits in (2, 4, 8)
    N = payload.shape[0]
    vpb = 8 // bits
    b = payload.to(torch.int64)[:, :, None]
    shifts = (torch.arange(vpb, device=payload.device, dtype=torch.int64) * bits)
    q = ((b >> shifts) & (2 ** bits - 1)).reshape(N, -1)[:, :num_feats].float()
    s = scale.float()
    inv = torch.where(s > 0, 1.0 / s.clamp(min=1e-30), torch.zeros_like(s))
    return q * inv[:, None] + rmin.float()[:, None]
