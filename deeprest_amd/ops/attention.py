"""Multi-head attention over the endpoint x time window.

The traffic encoder attends over the time axis of the encoded endpoint
window (north star: "multi-head attention over the endpoint x time window on
MFMA with LDS-staged tiles").  The forward is ONE fused HIP kernel per
(batch, head): QK^T on MFMA with K staged in LDS (XOR-swizzled), online
softmax in registers, PV on MFMA — no S x S score tensor is materialized.
It saves the per-row logsumexp so the backward can recompute P cheaply.

The backward is ONE fused kernel per (batch-head, 64-key tile) in a
TRANSPOSED layout — keys as MFMA rows, queries as columns — so S^T, dP^T and
dS^T share one C-fragment layout and the softmax-grad math is lane-local
(per-column lse and D-row); dK/dV accumulate in registers with exclusive
stores, dQ via fp32 atomics.  It recomputes P from the saved logsumexp
instead of storing the S x S score tensor (replaced a rocBLAS recompute-GEMM
chain: 1 ms -> 0.23 ms per step at the bench config).

Replaces and upgrades the reference's per-metric feature-mask "attention"
(reference: resource-estimation/qrnn.py:21-23,34).
"""

from __future__ import annotations

import math
from typing import Optional

import torch

from .native import require_native


def reference_mha(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
                  scale: Optional[float] = None) -> torch.Tensor:
    """q,k,v: (B, H, T, D). Plain composition (the numerics oracle)."""
    if scale is None:
        scale = 1.0 / math.sqrt(q.shape[-1])
    s = torch.matmul(q, k.transpose(-1, -2)) * scale
    p = torch.softmax(s.float(), dim=-1).to(q.dtype)
    return torch.matmul(p, v)


class _MHAFunction(torch.autograd.Function):
    @staticmethod
    def forward(ctx, q, k, v, scale):
        ext = require_native("mha_forward")
        o, lse = ext.mha_forward(q, k, v, float(scale))
        ctx.save_for_backward(q, k, v, o, lse)
        ctx.scale = float(scale)
        return o

    @staticmethod
    def backward(ctx, grad_o):
        ext = require_native("mha_backward")
        q, k, v, o, lse = ctx.saved_tensors
        dq, dk, dv = ext.mha_backward(q, k, v, o, grad_o.contiguous(), lse,
                                      ctx.scale)
        return dq.to(q.dtype), dk, dv, None


def mha_forward(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
                scale: Optional[float] = None) -> torch.Tensor:
    """Fused attention forward; q,k,v: (B, H, T, D) contiguous."""
    if scale is None:
        scale = 1.0 / math.sqrt(q.shape[-1])
    if q.is_cuda:
        return _MHAFunction.apply(q.contiguous(), k.contiguous(), v.contiguous(), scale)
    return reference_mha(q, k, v, scale)
