"""What-if trace synthesis.

Learns, per API endpoint (= root span's component_operation), the empirical
distribution over whole-trace call-path feature vectors, then synthesizes the
feature vector of a hypothetical traffic mix {api: count} by sampling trace
shapes with those weights and summing
(reference: resource-estimation/synthesizer.py:15-52).

Differences from the reference implementation (same capability, new design):
- shares one FeatureSpace with the featurizer instead of rebuilding its own;
- trace shapes are keyed by the bytes of their count vector (the reference
  round-trips through str()/eval());
- deterministic sampling via an explicit numpy Generator;
- `synthesize_series` produces a full (T, P) synthetic traffic matrix for the
  online what-if pipeline.
"""

from __future__ import annotations

from typing import Any, Dict, List, Optional, Sequence, Tuple

import numpy as np

from .featurize import FeatureSpace


class TraceSynthesizer:
    def __init__(self, feature_space: Optional[FeatureSpace] = None) -> None:
        self.feature_space = feature_space
        # api -> (stacked unique vectors (K, P), weights (K,))
        self.api2dist: Dict[str, Tuple[np.ndarray, np.ndarray]] = {}

    def fit(self, raw_data: Sequence[Dict[str, Any]]) -> "TraceSynthesizer":
        if self.feature_space is None:
            self.feature_space = FeatureSpace()
            for window in raw_data:
                for trace in window["traces"]:
                    self.feature_space.observe_trace(trace)
        fs = self.feature_space
        P = len(fs)

        counts: Dict[str, Dict[bytes, List[Any]]] = {}
        for window in raw_data:
            for trace in window["traces"]:
                api = f"{trace['component']}_{trace['operation']}"
                vec = np.zeros(P, dtype=np.int64)
                fs.count_trace(trace, vec)
                key = vec.tobytes()
                per_api = counts.setdefault(api, {})
                if key not in per_api:
                    per_api[key] = [vec, 0]
                per_api[key][1] += 1

        self.api2dist = {}
        for api, shapes in counts.items():
            vecs = np.stack([v for v, _ in shapes.values()])
            weights = np.asarray([n for _, n in shapes.values()], dtype=np.float64)
            self.api2dist[api] = (vecs, weights / weights.sum())
        return self

    @property
    def apis(self) -> List[str]:
        return list(self.api2dist.keys())

    def synthesize(
        self,
        expected_api_calls: Dict[str, int],
        rng: Optional[np.random.Generator] = None,
    ) -> np.ndarray:
        """Feature vector (P,) for one window of hypothetical traffic."""
        if rng is None:
            rng = np.random.default_rng()
        for api in expected_api_calls:
            if api not in self.api2dist:
                raise KeyError(f"API endpoint '{api}' does not exist.")
        P = len(self.feature_space)
        x = np.zeros(P, dtype=np.int64)
        for api, count in expected_api_calls.items():
            if count <= 0:
                continue
            vecs, weights = self.api2dist[api]
            picks = rng.choice(len(vecs), size=int(count), replace=True, p=weights)
            # sum sampled shape vectors; bincount avoids a python loop
            sel = np.bincount(picks, minlength=len(vecs))
            x += (sel[:, None] * vecs).sum(axis=0)
        return x

    def synthesize_series(
        self,
        traffic_plan: Sequence[Dict[str, int]],
        rng: Optional[np.random.Generator] = None,
    ) -> np.ndarray:
        """(T, P) synthetic traffic matrix for a per-window traffic plan."""
        if rng is None:
            rng = np.random.default_rng()
        return np.stack([self.synthesize(calls, rng) for calls in traffic_plan])
