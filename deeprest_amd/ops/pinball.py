"""Quantile (pinball) loss — fused elementwise + reduction HIP kernel.

Semantics follow the reference exactly (reference:
resource-estimation/qrnn.py:58-67): per metric m, per quantile q,

    e = label - pred_q;  l_q = max((q-1)*e, q*e)

summed over quantiles, averaged over (batch, time) per metric, then averaged
over metrics.  For equal-sized metrics this equals a weighted mean over all
elements, which is what the fused kernel computes in one pass (fwd) and one
elementwise pass (bwd).
"""

from __future__ import annotations

from typing import Sequence

import torch

from .native import require_native


def reference_pinball_loss(outputs: torch.Tensor, labels: torch.Tensor,
                           quantiles: Sequence[float]) -> torch.Tensor:
    """outputs: (B, T, M, Q); labels: (B, T, M)."""
    q = torch.as_tensor(list(quantiles), dtype=outputs.dtype, device=outputs.device)
    e = labels.unsqueeze(-1) - outputs                      # (B, T, M, Q)
    l = torch.maximum((q - 1.0) * e, q * e)                 # (B, T, M, Q)
    # sum over quantiles, mean over (B, T) per metric, mean over metrics
    return l.sum(dim=-1).mean(dim=(0, 1)).mean()


# device quantile tensors, cached per (device, quantiles): building one from a
# python list is a pageable H2D copy, which is both per-step overhead and
# forbidden inside hipGraph capture
_Q_CACHE: dict = {}


def _q_tensor(quantiles, device) -> torch.Tensor:
    key = (str(device), tuple(quantiles))
    q = _Q_CACHE.get(key)
    if q is None:
        q = torch.as_tensor(list(quantiles), dtype=torch.float32, device=device)
        _Q_CACHE[key] = q
    return q


class _PinballLoss(torch.autograd.Function):
    @staticmethod
    def forward(ctx, outputs, labels, quantiles):
        ext = require_native("pinball_loss")
        q = _q_tensor(quantiles, outputs.device)
        loss = ext.pinball_forward(outputs, labels, q)
        ctx.save_for_backward(outputs, labels, q)
        return loss

    @staticmethod
    def backward(ctx, grad_loss):
        ext = require_native("pinball_loss")
        outputs, labels, q = ctx.saved_tensors
        d_out = ext.pinball_backward(grad_loss, outputs, labels, q)
        return d_out, None, None


def pinball_loss(outputs: torch.Tensor, labels: torch.Tensor,
                 quantiles: Sequence[float] = (0.05, 0.50, 0.95)) -> torch.Tensor:
    if outputs.is_cuda:
        return _PinballLoss.apply(outputs.contiguous(), labels.contiguous(), tuple(quantiles))
    return reference_pinball_loss(outputs, labels, quantiles)
