"""The framework's single training step, shared by Trainer and bench.py.

One function owns the real step — autocast forward, pinball loss, backward
through the custom kernels, gradient all-reduce (DP over RCCL), fused Adam
update — so the driver-timed benchmark measures exactly the code path the
Trainer runs, not a parallel reimplementation (the reference's equivalent
loop: resource-estimation/estimate.py:65-75).

hipGraph capture plugs in here too: ``try_capture`` wraps the same step in a
GraphedTrainStep; batches matching the captured shape replay the graph,
everything else (tail batches, shape changes) falls back to the eager path.
"""

from __future__ import annotations

from typing import Callable, Optional

import torch


class TrainStep:
    def __init__(
        self,
        model: torch.nn.Module,
        optimizer,
        dist_ctx=None,
        loss_fn: Optional[Callable[[torch.Tensor, torch.Tensor], torch.Tensor]] = None,
        autocast_dtype: Optional[torch.dtype] = torch.bfloat16,
    ) -> None:
        self.model = model
        self.optimizer = optimizer
        self.dist = dist_ctx
        # bf16 outputs feed the pinball kernel directly (it computes in
        # fp32 registers; the gradient is sign-based quantile constants,
        # so no accuracy is lost and the ~280 MB/step output cast goes away)
        self.loss_fn = loss_fn or (lambda out, yb: model.loss(out, yb))
        self.autocast_dtype = autocast_dtype
        self._graphed = None

    # ------------------------------------------------------------- hipGraph
    def try_capture(self, x_example: torch.Tensor, y_example: torch.Tensor,
                    warmup: int = 3) -> bool:
        """Capture the whole step at this batch shape (single-process GPU
        runs only; GraphedTrainStep.build returns None otherwise).  The
        warmup replays are real optimizer steps on the example batch."""
        from .graphstep import GraphedTrainStep

        self._graphed = GraphedTrainStep.build(
            self.model, self.optimizer, self.loss_fn, x_example, y_example,
            autocast_dtype=self.autocast_dtype, warmup=warmup)
        return self._graphed is not None

    @property
    def graphed(self) -> bool:
        return self._graphed is not None

    # ----------------------------------------------------------------- step
    def __call__(self, xb: torch.Tensor, yb: torch.Tensor) -> torch.Tensor:
        """Run one optimizer step on (xb, yb); returns the on-device loss
        (no host sync — .item() on it is the caller's choice)."""
        g = self._graphed
        if g is not None and xb.shape == g.static_x.shape:
            return g.run(xb, yb)
        enabled = self.autocast_dtype is not None and xb.is_cuda
        with torch.autocast(
            device_type=xb.device.type,
            dtype=self.autocast_dtype or torch.bfloat16,
            enabled=enabled,
        ):
            out = self.model(xb)
            loss = self.loss_fn(out, yb)
        # with an active graph, grads must keep their captured addresses:
        # zero in place instead of dropping to None
        self.optimizer.zero_grad(set_to_none=self._graphed is None)
        loss.backward()
        if self.dist is not None:
            self.dist.all_reduce_gradients(self.model)
        self.optimizer.step()
        return loss
