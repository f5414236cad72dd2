"""Linear layer for huge-K weight gradients.

The per-resource quantile heads project (B*T*C, H') -> Q with B*T*C ~ 10^6:
the forward GEMM is fine, but autograd's dW = x^T grad is a (H', Q) output
with K ~ 10^6 — hipBLASLt picks a non-split-K tile and runs it on ~7
workgroups (measured 2.4 ms, 28% of the step).  This Function computes dW as
a bmm batched over the leading dim (hundreds of well-shaped GEMMs) and sums
in fp32 — same math, full chip.
"""

from __future__ import annotations

from typing import Optional

import torch


class _BigKLinear(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias):
        # x: (B, N, K); weight: (O, K); bias: (O,) or None
        w = weight.to(x.dtype)
        out = torch.matmul(x, w.t())
        if bias is not None:
            out = out + bias.to(out.dtype)
        ctx.save_for_backward(x, w)
        ctx.has_bias = bias is not None
        ctx.w_dtype = weight.dtype
        ctx.b_dtype = bias.dtype if bias is not None else None
        return out

    @staticmethod
    def backward(ctx, grad):
        x, w = ctx.saved_tensors
        g = grad.contiguous().to(x.dtype)
        dx = torch.matmul(g, w)
        # dW via leading-dim-batched GEMMs, fp32 accumulate of the partials
        dw = torch.bmm(g.transpose(1, 2), x).sum(0, dtype=torch.float32)
        db = grad.sum(dim=(0, 1), dtype=torch.float32) if ctx.has_bias else None
        return (dx, dw.to(ctx.w_dtype),
                db.to(ctx.b_dtype) if db is not None else None)


def bigk_linear(x: torch.Tensor, weight: torch.Tensor,
                bias: Optional[torch.Tensor] = None) -> torch.Tensor:
    """x: (..., K) with a large flattened batch; 3D-batches the dW GEMM."""
    shape = x.shape
    K = shape[-1]
    B = shape[0]
    x3 = x.reshape(B, -1, K)
    out = _BigKLinear.apply(x3, weight, bias)
    return out.reshape(*shape[:-1], weight.shape[0])
