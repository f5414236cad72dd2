"""LayerNorm (fwd/bwd) — hand-written HIP kernel on GPU.

Absent in the reference model; required by the new MHA traffic encoder
(SURVEY.md section 2.6 row "LayerNorm").  Memory-bound: the kernel does one
vectorized read (short4/float4), wave-level Welford-free two-pass-in-registers
reduction, and a fused affine, targeting the HBM roofline.
"""

from __future__ import annotations

import torch
import torch.nn.functional as F

from .native import require_native


def reference_layer_norm(x: torch.Tensor, weight: torch.Tensor, bias: torch.Tensor,
                         eps: float = 1e-5) -> torch.Tensor:
    return F.layer_norm(x, (x.shape[-1],), weight, bias, eps)


class _LayerNorm(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias, eps):
        ext = require_native("layer_norm")
        y, mean, rstd = ext.layer_norm_forward(x, weight, bias, float(eps))
        ctx.save_for_backward(x, weight, mean, rstd)
        return y

    @staticmethod
    def backward(ctx, grad_y):
        ext = require_native("layer_norm")
        x, weight, mean, rstd = ctx.saved_tensors
        dx, dw, db = ext.layer_norm_backward(grad_y.contiguous(), x, weight, mean, rstd)
        return dx, dw, db, None


def layer_norm(x: torch.Tensor, weight: torch.Tensor, bias: torch.Tensor,
               eps: float = 1e-5) -> torch.Tensor:
    if x.is_cuda:
        return _LayerNorm.apply(
            x.contiguous(), weight.float().contiguous(), bias.float().contiguous(), eps
        )
    return reference_layer_norm(x, weight, bias, eps)
