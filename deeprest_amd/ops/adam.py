"""Fused Adam step.

Replaces torch.optim.Adam's per-tensor eager update loop
(reference: resource-estimation/estimate.py:61,74) with ONE multi-tensor HIP
kernel per step on GPU: all parameter/grad/moment tensors are walked by a
single grid-stride kernel over a packed pointer table, so the optimizer costs
one launch regardless of parameter count.  CPU path uses torch._foreach ops.
"""

from __future__ import annotations

from typing import Iterable, List

import torch

from .native import require_native


@torch.no_grad()
def fused_adam_step(
    params: List[torch.Tensor],
    grads: List[torch.Tensor],
    exp_avgs: List[torch.Tensor],
    exp_avg_sqs: List[torch.Tensor],
    step: int,
    lr: float,
    beta1: float = 0.9,
    beta2: float = 0.999,
    eps: float = 1e-8,
    weight_decay: float = 0.0,
) -> None:
    if not params:
        return
    if params[0].is_cuda:
        ext = require_native("fused_adam_step")
        ext.fused_adam(params, grads, exp_avgs, exp_avg_sqs,
                       int(step), float(lr), float(beta1), float(beta2),
                       float(eps), float(weight_decay))
        return

    bias_c1 = 1.0 - beta1 ** step
    bias_c2 = 1.0 - beta2 ** step
    if weight_decay != 0.0:
        torch._foreach_add_(grads, params, alpha=weight_decay)
    torch._foreach_mul_(exp_avgs, beta1)
    torch._foreach_add_(exp_avgs, grads, alpha=1.0 - beta1)
    torch._foreach_mul_(exp_avg_sqs, beta2)
    torch._foreach_addcmul_(exp_avg_sqs, grads, grads, value=1.0 - beta2)
    denom = torch._foreach_sqrt(torch._foreach_div(exp_avg_sqs, bias_c2))
    torch._foreach_add_(denom, eps)
    step_size = lr / bias_c1
    torch._foreach_addcdiv_(params, exp_avgs, denom, value=-step_size)


class FusedAdam(torch.optim.Optimizer):
    """Optimizer wrapper over fused_adam_step (one kernel launch per step on GPU).

    ``capturable=True`` makes ``step()`` hipGraph-safe: the packed pointer
    table is cached on device (rebuilt only if any pointer changes, which can
    only happen outside capture) and the step counter is a device int32
    scalar incremented by a captured add — so a replayed training-step graph
    keeps exact Adam bias correction with zero host work.
    """

    def __init__(self, params: Iterable[torch.Tensor], lr: float = 1e-3,
                 betas=(0.9, 0.999), eps: float = 1e-8, weight_decay: float = 0.0,
                 capturable: bool = False):
        defaults = dict(lr=lr, betas=betas, eps=eps, weight_decay=weight_decay)
        super().__init__(params, defaults)
        if capturable and len(self.param_groups) > 1:
            raise ValueError("capturable FusedAdam supports a single param group")
        self.capturable = capturable
        self._step_t: torch.Tensor | None = None   # device int32 scalar
        self._lr_t: torch.Tensor | None = None     # device f32 scalar (lr)
        self._meta: torch.Tensor | None = None     # device int64 pointer table
        self._meta_key = None                      # pointer tuple behind _meta
        self._meta_total = 0
        self._meta_nt = 0

    def _capturable_step(self, params, grads, m, v, group) -> None:
        ext = require_native("fused_adam_step")
        dev = params[0].device
        key = tuple(t.data_ptr() for t in params + grads + m + v)
        if self._meta is None or self._meta_key != key:
            import itertools
            nt = len(params)
            ptrs = [t.data_ptr() for t in itertools.chain(params, grads, m, v)]
            cum, total = [], 0
            for p in params:
                total += p.numel()
                cum.append(total)
            self._meta = torch.tensor(ptrs + cum, dtype=torch.int64, device=dev)
            self._meta_key = key
            self._meta_total = total
            self._meta_nt = nt
        if self._step_t is None:
            # resume-consistent: host bookkeeping was already advanced for
            # this call, so seed the device counter one behind it
            start = int(self.state[params[0]]["step"]) - 1
            self._step_t = torch.full((1,), start, dtype=torch.int32, device=dev)
        if self._lr_t is None:
            # device-resident lr: captured replays read it, so LR schedules
            # work under hipGraph (set_lr updates it between replays)
            self._lr_t = torch.full((1,), float(group["lr"]),
                                    dtype=torch.float32, device=dev)
        self._step_t += 1  # device add: captured, so replays keep counting
        beta1, beta2 = group["betas"]
        ext.fused_adam_capturable(self._meta, self._meta_nt, self._meta_total,
                                  self._step_t, self._lr_t, beta1, beta2,
                                  group["eps"], group["weight_decay"])

    def set_lr(self, lr: float) -> None:
        """Schedule-safe lr update: refreshes the host groups AND the
        device scalar a captured (replayed) step reads."""
        for group in self.param_groups:
            group["lr"] = lr
        if self._lr_t is not None:
            self._lr_t.fill_(float(lr))

    def state_dict(self):
        # graph replays advance only the device counter; sync the host step
        # bookkeeping before checkpointing
        if self._step_t is not None:
            s = int(self._step_t.item())
            for st in self.state.values():
                if "step" in st:
                    st["step"] = s
        return super().state_dict()

    @torch.no_grad()
    def step(self, closure=None):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()
        for group in self.param_groups:
            params, grads, m, v = [], [], [], []
            for p in group["params"]:
                if p.grad is None:
                    continue
                state = self.state[p]
                if len(state) == 0:
                    state["step"] = 0
                    state["exp_avg"] = torch.zeros_like(p, memory_format=torch.preserve_format)
                    state["exp_avg_sq"] = torch.zeros_like(p, memory_format=torch.preserve_format)
                state["step"] += 1
                params.append(p)
                grads.append(p.grad)
                m.append(state["exp_avg"])
                v.append(state["exp_avg_sq"])
            if not params:
                continue
            if self.capturable and params[0].is_cuda:
                self._capturable_step(params, grads, m, v, group)
                continue
            step = self.state[params[0]]["step"]
            beta1, beta2 = group["betas"]
            fused_adam_step(params, grads, m, v, step, group["lr"], beta1, beta2,
                            group["eps"], group["weight_decay"])
        return loss
