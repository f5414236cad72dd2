"""Application sanity-checking via estimation residuals.

DeepRest's second capability (reference README.md:3): utilization that the
API traffic cannot justify — ransomware encryption, cryptojacking miners
(reference: locust/pow.py) — shows up as measured utilization persistently
above the traffic-conditioned upper quantile.  Score = normalized
exceedance over the q95 band, flagged when it persists.
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Dict, List

import numpy as np


@dataclass
class AnomalyReport:
    metric: str
    scores: np.ndarray            # (T,) per-window anomaly scores
    flags: np.ndarray             # (T,) bool
    windows: List[tuple]          # [(start, end), ...] flagged intervals

    @property
    def is_anomalous(self) -> bool:
        return bool(self.flags.any())


class AnomalyScorer:
    def __init__(self, threshold: float = 0.25, min_run: int = 3) -> None:
        """threshold: relative exceedance over the q95 band that counts;
        min_run: consecutive windows required before flagging (debounce)."""
        self.threshold = threshold
        self.min_run = min_run

    def score(self, measured: np.ndarray, q05: np.ndarray, q50: np.ndarray,
              q95: np.ndarray, metric: str = "") -> AnomalyReport:
        measured = np.asarray(measured, dtype=np.float64)
        band = np.maximum(np.asarray(q95) - np.asarray(q05), 1e-9)
        # how far above the upper quantile, in units of the predicted band
        exceed = (measured - np.asarray(q95)) / band
        scores = np.maximum(exceed, 0.0)
        over = scores > self.threshold
        flags = np.zeros_like(over)
        windows = []
        run = 0
        for i, o in enumerate(over):
            run = run + 1 if o else 0
            if run >= self.min_run:
                flags[i - run + 1 : i + 1] = True
        # contiguous flagged intervals
        in_run = False
        start = 0
        for i, f in enumerate(flags):
            if f and not in_run:
                in_run, start = True, i
            elif not f and in_run:
                in_run = False
                windows.append((start, i))
        if in_run:
            windows.append((start, len(flags)))
        return AnomalyReport(metric=metric, scores=scores, flags=flags,
                             windows=windows)

    def score_all(self, measured: Dict[str, np.ndarray],
                  preds: Dict[str, np.ndarray]) -> Dict[str, AnomalyReport]:
        """preds: {metric: (T, Q)} with Q = (q05, q50, q95)."""
        out = {}
        for name, series in measured.items():
            if name not in preds:
                continue
            p = np.asarray(preds[name])
            out[name] = self.score(series, p[:, 0], p[:, 1], p[:, 2], metric=name)
        return out
