"""Single-node data parallelism over RCCL / xGMI.

One process per GPU, ``torch.distributed`` with backend "nccl" (which IS
RCCL on ROCm).  The reference has no distributed compute at all (SURVEY.md
section 2.6/5.8); this module is the framework's scaling axis.

Gradient exchange strategy (xGMI-aware): each MI355X
has 7 point-to-point xGMI links (~153 GB/s each); ring all-reduce is bound by
one link, and for this model family gradients are small (about 1 MB for the
64-endpoint config), so latency dominates.  We therefore flatten ALL
gradients into ONE fused bucket per step (one collective, one launch) —
never the reference's nonexistent pattern nor torch DDP's 25 MB buckets.
For the 4096-endpoint config where the input-projection gradient grows to
hundreds of MB, the bucket splits at ``max_bucket_mb`` and the buckets
all-reduce asynchronously so copies overlap.

Testable without GPUs: backend "gloo" on CPU with world_size > 1 exercises
the identical code path (tests/test_dist_cpu.py).
"""

from __future__ import annotations

import os
from typing import List, Optional

import torch
import torch.distributed as dist


def is_distributed() -> bool:
    return dist.is_available() and dist.is_initialized()


def init_distributed(backend: Optional[str] = None,
                     device: Optional[torch.device] = None) -> Optional["DistContext"]:
    """Initialize from torchrun env vars; returns None when not launched
    distributed (WORLD_SIZE absent or 1)."""
    world_size = int(os.environ.get("WORLD_SIZE", "1"))
    if world_size <= 1:
        return None
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))
    if backend is None:
        # DEEPREST_DIST_BACKEND=gloo on a GPU box runs the whole distributed
        # path (sharded data, CUDA-tensor all-reduce, max-elapsed reduction)
        # through gloo — the only N>1 smoke possible on a 1-GPU lease, since
        # RCCL (like NCCL) refuses two ranks on one device ("Duplicate GPU
        # detected", measured: profiles/r02_dp_smoke.md)
        backend = os.environ.get("DEEPREST_DIST_BACKEND") or (
            "nccl" if torch.cuda.is_available() else "gloo")
    if torch.cuda.is_available() and device is None:
        # modulo lets an N-rank gloo smoke share the single GPU; with nccl
        # each rank must own a distinct device (driver scale runs)
        dev_index = local_rank % max(torch.cuda.device_count(), 1)
        torch.cuda.set_device(dev_index)
        device = torch.device("cuda", dev_index)
    elif device is None:
        device = torch.device("cpu")
    if not dist.is_initialized():
        dist.init_process_group(backend=backend)
    return DistContext(rank=rank, world_size=world_size,
                       local_rank=local_rank, device=device)


class DistContext:
    def __init__(self, rank: int, world_size: int, local_rank: int,
                 device: torch.device, max_bucket_mb: float = 128.0) -> None:
        self.rank = rank
        self.world_size = world_size
        self.local_rank = local_rank
        self.device = device
        self.max_bucket_bytes = int(max_bucket_mb * 1024 * 1024)
        self._flat_buf: Optional[torch.Tensor] = None

    # --------------------------------------------------------------- helpers
    def barrier(self) -> None:
        if is_distributed():
            dist.barrier()

    def broadcast_parameters(self, model: torch.nn.Module) -> None:
        """Rank-0 weights to all ranks (replaces seed-coupling assumptions)."""
        if not is_distributed():
            return
        with torch.no_grad():
            for p in model.state_dict().values():
                if isinstance(p, torch.Tensor):
                    dist.broadcast(p, src=0)

    # --------------------------------------------------- gradient all-reduce
    def all_reduce_gradients(self, model: torch.nn.Module) -> None:
        if not is_distributed():
            return
        grads: List[torch.Tensor] = [
            p.grad for p in model.parameters() if p.grad is not None
        ]
        if not grads:
            return
        elem = grads[0].element_size()
        total_bytes = sum(g.numel() for g in grads) * elem

        if total_bytes <= self.max_bucket_bytes:
            # ONE fused bucket: a single collective per step (latency-bound regime)
            numel = sum(g.numel() for g in grads)
            if self._flat_buf is None or self._flat_buf.numel() < numel \
                    or self._flat_buf.dtype != grads[0].dtype:
                self._flat_buf = torch.empty(
                    numel, dtype=grads[0].dtype, device=grads[0].device
                )
            flat = self._flat_buf[:numel]
            off = 0
            for g in grads:
                flat[off : off + g.numel()].copy_(g.view(-1))
                off += g.numel()
            dist.all_reduce(flat, op=dist.ReduceOp.SUM)
            flat.div_(self.world_size)
            off = 0
            for g in grads:
                g.view(-1).copy_(flat[off : off + g.numel()])
                off += g.numel()
            return

        # large model: size-bounded buckets, async so transfers overlap
        handles = []
        bucket: List[torch.Tensor] = []
        bucket_bytes = 0

        def flush():
            nonlocal bucket, bucket_bytes
            if not bucket:
                return
            flat = torch.cat([g.view(-1) for g in bucket])
            h = dist.all_reduce(flat, op=dist.ReduceOp.SUM, async_op=True)
            handles.append((h, flat, list(bucket)))
            bucket = []
            bucket_bytes = 0

        for g in grads:
            nbytes = g.numel() * elem
            if bucket_bytes + nbytes > self.max_bucket_bytes:
                flush()
            bucket.append(g)
            bucket_bytes += nbytes
        flush()

        for h, flat, bucket_grads in handles:
            h.wait()
            flat.div_(self.world_size)
            off = 0
            for g in bucket_grads:
                g.view(-1).copy_(flat[off : off + g.numel()])
                off += g.numel()

    # ------------------------------------------------------------ reductions
    def all_reduce_scalar(self, value: float, op: str = "sum") -> float:
        if not is_distributed():
            return value
        t = torch.tensor([value], dtype=torch.float64,
                         device=self.device if self.device.type == "cuda" else "cpu")
        dist.all_reduce(t, op=dist.ReduceOp.MAX if op == "max" else dist.ReduceOp.SUM)
        return float(t.item())
