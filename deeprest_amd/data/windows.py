"""Windowing and normalization utilities.

Matches the reference's semantics:
- sliding_window: stride-1 windows, the last (len-window_size) starts only
  (reference: resource-estimation/utils.py:4-5);
- min-max normalization fit on the train split only
  (reference: resource-estimation/qrnn.py:69-75).

Implemented with numpy stride tricks (no python-loop copy) and an explicit
scaler object so the fitted (scale, min) pairs can be checkpointed — in the
reference they live only in process memory.
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Tuple

import numpy as np


def sliding_window(ts: np.ndarray, window_size: int) -> np.ndarray:
    """Stride-1 sliding windows: output[i] = ts[i:i+window_size].

    Output length is len(ts) - window_size (the final full window starting at
    len-window_size is excluded, matching reference utils.py:4-5).
    Returns a contiguous copy of shape (N, window_size, *ts.shape[1:]).
    """
    ts = np.ascontiguousarray(ts)
    n = len(ts) - window_size
    if n <= 0:
        return np.empty((0, window_size) + ts.shape[1:], dtype=ts.dtype)
    # explicit per-window slice copies: each is one contiguous memcpy, so the
    # copy runs at memory bandwidth regardless of how numpy would iterate the
    # overlapping strided view
    out = np.empty((n, window_size) + ts.shape[1:], dtype=ts.dtype)
    for i in range(n):
        out[i] = ts[i : i + window_size]
    return out


def minmax_fit(M: np.ndarray, split: int) -> Tuple[float, float]:
    """(min, max) over the first `split` entries (train portion)."""
    train = M[:split]
    return float(np.min(train)), float(np.max(train))


def minmax_apply(M: np.ndarray, min_val: float, max_val: float) -> np.ndarray:
    rng = max_val - min_val
    if rng == 0.0:
        return M
    return (M - min_val) / rng


@dataclass
class MinMaxScaler:
    """Checkpointable min-max scaler (fit on train split only)."""

    min_val: float = 0.0
    max_val: float = 1.0

    @property
    def scale(self) -> float:
        return self.max_val - self.min_val

    def fit(self, M: np.ndarray, split: int) -> "MinMaxScaler":
        self.min_val, self.max_val = minmax_fit(M, split)
        return self

    def transform(self, M: np.ndarray) -> np.ndarray:
        return minmax_apply(M, self.min_val, self.max_val)

    def fit_transform(self, M: np.ndarray, split: int) -> np.ndarray:
        return self.fit(M, split).transform(M)

    def inverse_transform(self, M: np.ndarray) -> np.ndarray:
        if self.scale == 0.0:
            return M
        return M * self.scale + self.min_val

    def state_dict(self) -> dict:
        return {"min_val": self.min_val, "max_val": self.max_val}

    @staticmethod
    def from_state_dict(state: dict) -> "MinMaxScaler":
        return MinMaxScaler(min_val=state["min_val"], max_val=state["max_val"])
