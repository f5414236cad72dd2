"""Dynamic micro-batching for the prediction service.

Concurrent requests coalesce into ONE predictor call — on GPU that is one
hipGraph replay instead of one per request (the captured path's win grows
with batch size, profiles/r02_serve_p50.md).  Synchronous, thread-safe
interface: FastAPI runs sync endpoints on a thread pool, so concurrent
HTTP requests naturally meet here.

Policy: a batch flushes when the pending window count reaches
``max_batch`` or ``max_wait_ms`` after the first request arrived —
the classic latency/throughput knob.
"""

from __future__ import annotations

import threading
from typing import Dict, List, Optional

import numpy as np
import torch


class MicroBatcher:
    def __init__(self, predictor, max_batch: int = 1024,
                 max_wait_ms: float = 2.0, timeout_s: float = 30.0) -> None:
        self.predictor = predictor
        self.max_batch = max_batch
        self.max_wait_ms = max_wait_ms
        self.timeout_s = timeout_s
        self._lock = threading.Lock()
        # one flush at a time: the Predictor's graph buffers are shared
        # state (two overlapping flushes raced into concurrent graph
        # capture); batches still COALESCE freely, execution serializes —
        # the correct discipline for one GPU
        self._run_lock = threading.Lock()
        self._pending: List[tuple] = []        # (windows, event, slot)
        self._timer: Optional[threading.Timer] = None
        self.batches_run = 0                   # observability
        self.requests_served = 0

    # ------------------------------------------------------------ internal
    def _drain_locked(self) -> List[tuple]:
        batch, self._pending = self._pending, []
        if self._timer is not None:
            self._timer.cancel()
            self._timer = None
        return batch

    def _flush_timer(self) -> None:
        with self._lock:
            batch = self._drain_locked()
        if batch:
            self._run(batch)

    def _run(self, batch: List[tuple]) -> None:
        with self._run_lock:
            self._run_locked(batch)

    def _run_locked(self, batch: List[tuple]) -> None:
        try:
            if isinstance(batch[0][0], torch.Tensor):
                # device tensors (each client thread already did its H2D):
                # concatenate ON DEVICE and take the tensor fast path
                out = self.predictor.predict_tensor(
                    torch.cat([w for w, _, _ in batch], dim=0))
            else:
                out = self.predictor.predict(np.concatenate(
                    [np.asarray(w, dtype=np.float64) for w, _, _ in batch],
                    axis=0))
            err = None
        except Exception as exc:  # noqa: BLE001 — deliver to every waiter
            out, err = None, exc
        s = 0
        for w, ev, slot in batch:
            n = len(w)
            if err is None:
                slot["out"] = {k: v[s : s + n] for k, v in out.items()}
            else:
                slot["err"] = err
            s += n
            ev.set()
        self.batches_run += 1
        self.requests_served += len(batch)

    # ------------------------------------------------------------- public
    def predict(self, traffic_windows: np.ndarray) -> Dict[str, np.ndarray]:
        """Blocking predict; concurrent callers share one predictor call.

        When the predictor exposes the device fast path, this thread does
        its own H2D copy up front (releases the GIL) so enqueue cost
        parallelizes across client threads and the flush is one device-side
        concat + one graph replay."""
        if hasattr(self.predictor, "predict_tensor"):
            x = np.ascontiguousarray(np.asarray(traffic_windows),
                                     dtype=np.float32)
            w = torch.from_numpy(x).to(self.predictor.device)
        else:
            w = np.asarray(traffic_windows, dtype=np.float64)
        ev = threading.Event()
        slot: dict = {}
        run_now: Optional[List[tuple]] = None
        with self._lock:
            self._pending.append((w, ev, slot))
            n_tot = sum(len(x) for x, _, _ in self._pending)
            if n_tot >= self.max_batch:
                run_now = self._drain_locked()
            elif self._timer is None:
                self._timer = threading.Timer(self.max_wait_ms / 1000.0,
                                              self._flush_timer)
                self._timer.daemon = True
                self._timer.start()
        if run_now:
            self._run(run_now)
        if not ev.wait(self.timeout_s):
            raise TimeoutError("micro-batch flush did not complete in time")
        if "err" in slot:
            raise slot["err"]
        return slot["out"]
