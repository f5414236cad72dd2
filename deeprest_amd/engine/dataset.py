"""Windowed train/test dataset with checkpointable normalization.

Reproduces the reference's data preparation exactly
(reference: resource-estimation/estimate.py:22-57):

- stride-1 sliding windows of length ``step_size`` over the traffic matrix
  and each resource series;
- ``split = int(len(X) * split_fraction)`` train/test boundary;
- global min-max normalization of X and per-metric min-max of y, fit on the
  train split only (qrnn.py:69-75), with the fitted (scale, min) pairs kept
  as state so inference is reproducible from a checkpoint (the reference
  keeps them only in process memory).
"""

from __future__ import annotations

from typing import List

import numpy as np
import torch

from ..data.featurize import FeaturizedData
from ..data.windows import MinMaxScaler, sliding_window


class EstimationDataset:
    def __init__(
        self,
        data: FeaturizedData,
        step_size: int = 60,
        split_fraction: float = 0.40,
        target_transform: str = "none",
    ) -> None:
        """``target_transform="log1p"``: fit scalers on log1p(y) and invert
        with expm1 at denormalization.  Monotone, so quantile outputs stay
        quantiles; compresses the unseen-scale extrapolation range (a 3x
        traffic query moves the target by +log 3 instead of 3x past the
        fitted min-max — the measured failure mode of the bounded net on
        the reference's unseen-scale axis, profiles/r02_unseen_traffic.md)."""
        self.data = data
        self.step_size = step_size
        self.target_transform = target_transform
        self.metric_names = data.metric_names

        # The traffic windows are the BIG tensor at 4096-endpoint scale
        # ((N, 60, 205k): ~9 GB as f32) — build them as float32 once and
        # normalize IN PLACE instead of the naive f64 window + transform +
        # cast chain (3 x 17.7 GB of transients, measured 151 s -> ~7 s).
        # Counts are integers < 2^24, so the f32 min/max fit is exact and the
        # scaler state matches the f64 formulation bit-for-bit.
        traffic = np.asarray(data.traffic, dtype=np.float32)
        y_flat = np.stack([data.resources[n] for n in self.metric_names], axis=-1)

        X = sliding_window(traffic, step_size)                 # (N, T, P) f32
        y = sliding_window(y_flat.astype(np.float64), step_size)  # (N, T, M)
        self.num_windows = len(X)
        self.split = int(self.num_windows * split_fraction)
        if self.split <= 0 or self.split >= self.num_windows:
            raise ValueError(
                f"split {self.split} out of range for {self.num_windows} windows"
            )

        # raw (denormalized) labels, needed by the baselines and error eval
        self.y_raw = y.copy()
        if target_transform == "log1p":
            y = np.log1p(np.maximum(y, 0.0))
        elif target_transform != "none":
            raise ValueError(f"unknown target_transform {target_transform!r}")

        self.x_scaler = MinMaxScaler().fit(X, self.split)
        if self.x_scaler.scale != 0.0:
            X -= np.float32(self.x_scaler.min_val)
            X *= np.float32(1.0 / self.x_scaler.scale)
        self.y_scalers: List[MinMaxScaler] = []
        for idx in range(y.shape[-1]):
            sc = MinMaxScaler().fit(y[:, :, idx], self.split)
            y[:, :, idx] = sc.transform(y[:, :, idx])
            self.y_scalers.append(sc)

        self.X = torch.from_numpy(X)                           # already f32 contiguous
        self.y = torch.from_numpy(np.ascontiguousarray(y, dtype=np.float32))

    # ------------------------------------------------------------------ views
    @property
    def X_train(self) -> torch.Tensor:
        return self.X[: self.split]

    @property
    def y_train(self) -> torch.Tensor:
        return self.y[: self.split]

    @property
    def X_test(self) -> torch.Tensor:
        return self.X[self.split :]

    @property
    def y_test(self) -> torch.Tensor:
        return self.y[self.split :]

    def eval_window_indices(self, max_cycles: int = 9) -> List[int]:
        """Non-overlapping test windows: iv % step == 0, up to max_cycles
        (reference: estimate.py:85-88)."""
        out = []
        n_test = self.num_windows - self.split
        for iv in range(n_test):
            if iv % self.step_size == 0:
                out.append(iv)
                if len(out) >= max_cycles:
                    break
        return out

    def denormalize_metric(self, values: np.ndarray, idx: int) -> np.ndarray:
        out = self.y_scalers[idx].inverse_transform(values)
        if self.target_transform == "log1p":
            out = np.expm1(out)
        return out

    # ------------------------------------------------------------- checkpoint
    def scaler_state(self) -> dict:
        return {
            "x_scaler": self.x_scaler.state_dict(),
            "y_scalers": [s.state_dict() for s in self.y_scalers],
            "metric_names": self.metric_names,
            "step_size": self.step_size,
            "split": self.split,
            "target_transform": self.target_transform,
        }
