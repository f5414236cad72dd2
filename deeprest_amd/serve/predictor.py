"""Online prediction path.

Loads a checkpoint (weights + spec + fitted scalers + call-path feature
space M — everything inference needs) and serves
batched quantile predictions.  On GPU the forward is captured once into a
hipGraph (torch.cuda.CUDAGraph IS hipGraph on ROCm) at a fixed batch shape;
smaller requests are padded and replayed — one graph launch instead of
hundreds of kernel launches per request (north star: "the online prediction
step is hipGraph-captured").
"""

from __future__ import annotations

from typing import Dict, List, Optional

import numpy as np
import torch

from ..data.featurize import FeatureSpace
from ..data.windows import MinMaxScaler
from ..engine.checkpoint import load_checkpoint
from ..models.net import DeepRestNet


class Predictor:
    def __init__(
        self,
        model: DeepRestNet,
        x_scaler: MinMaxScaler,
        y_scalers: List[MinMaxScaler],
        metric_names: List[str],
        feature_space: Optional[FeatureSpace] = None,
        device: Optional[torch.device] = None,
        graph_batch: int = 64,
        use_graph: bool = True,
    ) -> None:
        self.device = device or torch.device(
            "cuda" if torch.cuda.is_available() else "cpu"
        )
        self.model = model.to(self.device).eval()
        self.x_scaler = x_scaler
        self.y_scalers = y_scalers
        self.metric_names = metric_names
        self.feature_space = feature_space
        self.graph_batch = graph_batch
        self.use_graph = use_graph and self.device.type == "cuda"
        self._graph = None
        self._graph_in: Optional[torch.Tensor] = None
        self._graph_out: Optional[torch.Tensor] = None
        self._graph_T: Optional[int] = None

    @staticmethod
    def from_checkpoint(path: str, device: Optional[torch.device] = None,
                        **kw) -> "Predictor":
        state = load_checkpoint(path)
        model = DeepRestNet.from_full_state(state["model"])
        sc = state["scalers"]
        x_scaler = MinMaxScaler.from_state_dict(sc["x_scaler"])
        y_scalers = [MinMaxScaler.from_state_dict(s) for s in sc["y_scalers"]]
        fs = (FeatureSpace.from_state_dict(state["feature_space"])
              if state.get("feature_space") else None)
        return Predictor(model, x_scaler, y_scalers, sc["metric_names"],
                         feature_space=fs, device=device, **kw)

    # ---------------------------------------------------------------- capture
    def _ensure_graph(self, T: int, P: int) -> None:
        if self._graph is not None and self._graph_T == T:
            return
        B = self.graph_batch
        self._graph_in = torch.zeros(B, T, P, device=self.device)
        # warmup on a side stream (required before capture)
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for _ in range(2):
                with torch.no_grad():
                    self.model(self._graph_in)
        torch.cuda.current_stream().wait_stream(s)
        g = torch.cuda.CUDAGraph()
        with torch.cuda.graph(g):
            with torch.no_grad():
                self._graph_out = self.model(self._graph_in)
        self._graph = g
        self._graph_T = T

    # ---------------------------------------------------------------- predict
    @torch.no_grad()
    def predict_normalized(self, x: torch.Tensor) -> torch.Tensor:
        """x: (N, T, P) normalized traffic -> (N, T, M, Q).

        Adaptive: requests up to graph_batch replay the captured hipGraph
        (measured 17% lower p50 on small latency-sensitive requests); bulk
        requests beyond it run one big eager batch (measured faster than
        chunked replays at 1k windows)."""
        x = x.to(self.device, dtype=torch.float32)
        N, T, P = x.shape
        if not self.use_graph or N > self.graph_batch:
            return self.model(x)
        self._ensure_graph(T, P)
        outs = []
        B = self.graph_batch
        for s in range(0, N, B):
            chunk = x[s : s + B]
            n = chunk.shape[0]
            self._graph_in[:n].copy_(chunk)
            if n < B:
                self._graph_in[n:].zero_()
            self._graph.replay()
            outs.append(self._graph_out[:n].clone())
        return torch.cat(outs, dim=0)

    def predict(self, traffic_windows: np.ndarray) -> Dict[str, np.ndarray]:
        """Raw call-path count windows (N, T, P) -> per-metric denormalized
        quantile predictions {metric: (N, T, Q)}."""
        x = np.asarray(traffic_windows, dtype=np.float64)
        xn = self.x_scaler.transform(x)
        out = self.predict_normalized(torch.from_numpy(xn).float())
        out = out.float().cpu().numpy()            # (N, T, M, Q)
        # quantile regression can emit crossed quantiles (q95 < q50) early in
        # training; serving consumers (anomaly bands, demo plots) assume a
        # monotone triple, so sort the Q axis — a no-op once calibrated
        out = np.sort(out, axis=-1)
        preds = {}
        for m, name in enumerate(self.metric_names):
            preds[name] = np.maximum(
                self.y_scalers[m].inverse_transform(out[:, :, m, :]), 1e-6
            )
        return preds

    def predict_what_if(self, synthesizer, traffic_plan, step_size: int,
                        rng=None) -> Dict[str, np.ndarray]:
        """What-if pipeline: synthesize traffic for a hypothetical API mix,
        window it, and predict (reference: synthesizer.py + SURVEY.md 3.3)."""
        series = synthesizer.synthesize_series(traffic_plan, rng=rng)  # (T, P)
        n = len(series) - step_size
        if n <= 0:
            windows = series[None, :, :].astype(np.float64)
        else:
            from ..data.windows import sliding_window

            windows = sliding_window(series.astype(np.float64), step_size)
        return self.predict(windows)
