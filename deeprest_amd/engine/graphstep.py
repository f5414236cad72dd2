"""hipGraph-captured training step.

The full training step — autocast forward, pinball loss, backward through
every custom kernel, and the fused Adam update — is ~200 kernel launches on
static shapes; capturing it once into a hipGraph (torch.cuda.CUDAGraph IS
hipGraph on ROCm) and replaying per step removes the per-launch host
overhead and the gaps between launches.  The reference has no equivalent
(its training loop is eager PyTorch; reference: resource-estimation/
estimate.py:61-76); this is the MI355X-native "capture launch-bound inner
loops in hipGraphs" rule applied to training, not just serving.

Capture-safety in this codebase:
 - pinball backward reads the upstream grad on-device (csrc/pinball.hip);
 - FusedAdam(capturable=True) keeps its pointer table and step counter on
   device (ops/adam.py), so bias correction stays exact across replays;
 - the model's custom autograd Functions do only device work in backward.

Use ``GraphedTrainStep.build(...)``; it returns None (caller stays eager)
when capture is unsupported or fails — capture is an optimization, never a
correctness requirement.

Contract for callers: do NOT host-synchronize between consecutive replays
(a `torch.cuda.synchronize()` after every replay corrupts training on this
stack a few dozen replays in — measured and isolated in
profiles/r02_graph_capture_notes.md; epoch-granularity syncs are fine and
are what Trainer/bench do).
"""

from __future__ import annotations

import os
from typing import Callable, Optional

import torch


class GraphedTrainStep:
    """Replays one captured (forward + loss + backward + optimizer) step.

    ``loss_fn(output, target) -> scalar loss`` runs inside the capture.
    Inputs are copied into static buffers each step; the returned loss is the
    static device scalar (clone it before the next replay if it must persist).
    """

    def __init__(self, model: torch.nn.Module, optimizer,
                 loss_fn: Callable[[torch.Tensor, torch.Tensor], torch.Tensor],
                 x_example: torch.Tensor, y_example: torch.Tensor,
                 autocast_dtype: Optional[torch.dtype] = torch.bfloat16,
                 warmup: int = 3, static_inputs: bool = False, pool=None):
        """``static_inputs=True``: capture directly over the PASSED tensors
        (caller keeps them alive and stable) — a replay then reads them
        zero-copy, e.g. a persistent slice of the resident training set.
        ``pool``: share one capture memory pool across several graphs
        (multi-offset cycling) instead of one activation arena each."""
        self.model = model
        self.optimizer = optimizer
        self.loss_fn = loss_fn
        self.autocast_dtype = autocast_dtype
        self.static_x = x_example if static_inputs else x_example.clone()
        self.static_y = y_example if static_inputs else y_example.clone()
        self._pool = pool

        # warmup on a side stream: allocates grads, Adam state, the fused
        # all-reduce bucket, autograd workspace — everything whose pointers
        # the captured graph will bake in.  Capture happens on the SAME
        # stream: autograd binds each parameter's AccumulateGrad node to the
        # stream it first ran on, so warming up on one stream and capturing
        # on another records cross-stream edges that replay unsafely
        # (measured: replays went NaN when the host synchronized between
        # replays — tools/probe_debug3.py)
        side = torch.cuda.Stream()
        side.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(side):
            for _ in range(max(warmup, 2)):
                self._one_step()
        torch.cuda.current_stream().wait_stream(side)

        self.graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self.graph, stream=side, pool=self._pool):
            self.static_loss = self._one_step()

    def _one_step(self) -> torch.Tensor:
        enabled = self.autocast_dtype is not None
        # cache_enabled=False is REQUIRED under graph capture: with the
        # weight-cast cache on, captured replays hold references into cache
        # entries that any later EAGER autocast step (e.g. a ragged tail
        # batch) invalidates — measured as deterministic NaN corruption of
        # every parameter a few epochs in (tools/probe_debug3.py: drop-last
        # training was clean, any eager tail step poisoned the replays)
        with torch.autocast(device_type="cuda",
                            dtype=self.autocast_dtype or torch.bfloat16,
                            enabled=enabled, cache_enabled=False):
            out = self.model(self.static_x)
            loss = self.loss_fn(out, self.static_y)
        # set_to_none=False: grads must stay at fixed addresses across replays
        self.optimizer.zero_grad(set_to_none=False)
        loss.backward()
        self.optimizer.step()
        return loss

    def run(self, x: torch.Tensor, y: torch.Tensor) -> torch.Tensor:
        if x.data_ptr() != self.static_x.data_ptr():
            self.static_x.copy_(x, non_blocking=True)
        if y.data_ptr() != self.static_y.data_ptr():
            self.static_y.copy_(y, non_blocking=True)
        self.graph.replay()
        return self.static_loss

    def replay(self) -> torch.Tensor:
        """Zero-copy replay over the captured (static) inputs."""
        self.graph.replay()
        return self.static_loss

    @staticmethod
    def build(model, optimizer, loss_fn, x_example, y_example,
              autocast_dtype=torch.bfloat16, warmup: int = 3
              ) -> Optional["GraphedTrainStep"]:
        """Capture if possible; None (stay eager) if capture is unavailable.

        Capture is restricted to single-process runs: collectives inside a
        captured graph are left to the eager path until RCCL graph capture
        has been validated on the target pool.
        """
        if not torch.cuda.is_available():
            return None
        if os.environ.get("DEEPREST_NO_GRAPH", "0") == "1":
            return None
        if torch.distributed.is_available() and torch.distributed.is_initialized():
            return None
        try:
            return GraphedTrainStep(model, optimizer, loss_fn,
                                    x_example, y_example,
                                    autocast_dtype=autocast_dtype, warmup=warmup)
        except Exception as exc:  # capture failure -> eager fallback
            import warnings
            warnings.warn(f"hipGraph train-step capture failed ({exc}); "
                          "falling back to eager stepping")
            return None
