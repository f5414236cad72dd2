"""Online prediction path.

Loads a checkpoint (weights + spec + fitted scalers + call-path feature
space M — everything inference needs) and serves
batched quantile predictions.  On GPU the forward is captured once into a
hipGraph (torch.cuda.CUDAGraph IS hipGraph on ROCm) per configured batch
size; every request is served by replays of the smallest graph that fits —
one graph launch instead of hundreds of kernel launches per request (north
star: "the online prediction step is hipGraph-captured", BASELINE config 3:
batched 1k-window inference).

Staging-buffer ingestion: requests are copied straight into the captured
graph's fixed input buffer (H2D for numpy requests, D2D for resident
tensors) — no intermediate device tensor, so bulk 1k-window inference is a
single copy + one replay, which beats the eager path (round 1's single
64-window graph lost at bulk sizes because 1k windows took 16 chunked
replays; measured in profiles/r02_serve_p50.md).
"""

from __future__ import annotations

from typing import Dict, List, Optional, Sequence, Tuple

import numpy as np
import torch

from ..data.featurize import FeatureSpace
from ..data.windows import MinMaxScaler
from ..engine.checkpoint import load_checkpoint
from ..models.net import DeepRestNet

DEFAULT_GRAPH_BATCHES = (16, 256, 1024)


class Predictor:
    def __init__(
        self,
        model: DeepRestNet,
        x_scaler: MinMaxScaler,
        y_scalers: List[MinMaxScaler],
        metric_names: List[str],
        feature_space: Optional[FeatureSpace] = None,
        device: Optional[torch.device] = None,
        graph_batch: Optional[int] = None,
        graph_batches: Optional[Sequence[int]] = None,
        use_graph: bool = True,
        target_transform: str = "none",
        residual_ridge: Optional[np.ndarray] = None,   # (P+1, M): the net's
        # outputs are residuals over this ridge (train.residual_base)
        conformal: Optional[np.ndarray] = None,        # (M,) CQR band widening
    ) -> None:
        self.target_transform = target_transform
        self.residual_ridge = (np.asarray(residual_ridge)
                               if residual_ridge is not None else None)
        self.conformal = (np.asarray(conformal)
                          if conformal is not None else None)
        self._ridge_t: Optional[torch.Tensor] = None    # device caches
        self._conformal_t: Optional[torch.Tensor] = None
        self.device = device or torch.device(
            "cuda" if torch.cuda.is_available() else "cpu"
        )
        self.model = model.to(self.device).eval()
        self.x_scaler = x_scaler
        self.y_scalers = y_scalers
        self.metric_names = metric_names
        self.feature_space = feature_space
        if graph_batches is None:
            graph_batches = (graph_batch,) if graph_batch else DEFAULT_GRAPH_BATCHES
        self.graph_batches: Tuple[int, ...] = tuple(sorted(set(graph_batches)))
        self.use_graph = use_graph and self.device.type == "cuda"
        # (batch, T) -> (graph, input buffer, output buffer)
        self._graphs: Dict[Tuple[int, int], tuple] = {}

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
        kw.setdefault("target_transform", sc.get("target_transform", "none"))
        kw.setdefault("residual_ridge",
                      (state.get("extra") or {}).get("residual_ridge"))
        kw.setdefault("conformal", (state.get("extra") or {}).get("conformal"))
        return Predictor(model, x_scaler, y_scalers, sc["metric_names"],
                         feature_space=fs, device=device, **kw)

    @property
    def captured_batches(self) -> List[int]:
        return sorted({b for (b, _t) in self._graphs})

    # ---------------------------------------------------------------- capture
    def _graph_for(self, n: int, T: int, P: int) -> tuple:
        """Smallest configured graph batch >= n (n must be <= max batch)."""
        b = next(bb for bb in self.graph_batches if bb >= n)
        key = (b, T)
        got = self._graphs.get(key)
        if got is not None:
            return got
        gin = torch.zeros(b, T, P, device=self.device)
        # warmup on a side stream (required before capture)
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for _ in range(2):
                with torch.no_grad():
                    self.model(gin)
        torch.cuda.current_stream().wait_stream(s)
        g = torch.cuda.CUDAGraph()
        with torch.cuda.graph(g, stream=s):
            with torch.no_grad():
                gout = self.model(gin)
        got = (g, gin, gout)
        self._graphs[key] = got
        return got

    # ----------------------------------------------------- staged ingestion
    def staging_buffer(self, n: int, T: int, P: int) -> torch.Tensor:
        """A writable (n, T, P) view of the captured graph's input buffer.

        Streamed ingestion writes normalized windows directly here (no
        intermediate tensor), then calls ``predict_staged(n)`` — the
        whole-request cost is one replay."""
        _, gin, _ = self._graph_for(n, T, P)
        return gin[:n]

    @torch.no_grad()
    def predict_staged(self, n: int, T: int) -> torch.Tensor:
        """Replay the graph whose staging buffer was filled with n windows."""
        b = next(bb for bb in self.graph_batches if bb >= n)
        g, gin, gout = self._graphs[(b, T)]
        if n < b:
            gin[n:].zero_()
        g.replay()
        return gout[:n]

    # ---------------------------------------------------------------- predict
    @torch.no_grad()
    def predict_normalized(self, x: torch.Tensor) -> torch.Tensor:
        """x: (N, T, P) normalized traffic -> (N, T, M, Q).

        Requests up to the largest configured graph batch are one padded
        replay; larger requests chunk at the largest batch (a 4096-window
        request = four 1024-replays).  CPU inputs are copied H2D straight
        into the staging buffer."""
        x = x.float()
        N, T, P = x.shape
        if not self.use_graph:
            return self.model(x.to(self.device))
        bmax = self.graph_batches[-1]
        outs = []
        s = 0
        while s < N:
            n = min(N - s, bmax)
            g, gin, gout = self._graph_for(n, T, P)
            gin[:n].copy_(x[s : s + n])
            if n < gin.shape[0]:
                gin[n:].zero_()
            g.replay()
            outs.append(gout[:n].clone())
            s += n
        return outs[0] if len(outs) == 1 else torch.cat(outs, dim=0)

    def predict(self, traffic_windows: np.ndarray) -> Dict[str, np.ndarray]:
        """Raw call-path count windows (N, T, P) -> per-metric denormalized
        quantile predictions {metric: (N, T, Q)}.

        The whole request pipeline (normalization, residual base, quantile
        sort, conformal widening) runs on-device in f32: the former f64
        numpy normalization alone cost ~50 ms per 16-window request at the
        256-endpoint width and GIL-serialized concurrent serving."""
        x = np.ascontiguousarray(np.asarray(traffic_windows),
                                 dtype=np.float32)
        return self.predict_tensor(torch.from_numpy(x).to(self.device))

    @torch.no_grad()
    def predict_tensor(self, xt: torch.Tensor) -> Dict[str, np.ndarray]:
        """Raw count windows already on device as f32 (N, T, P) — the
        zero-extra-copy entry the micro-batcher uses (client threads do
        their own H2D, the batch concatenates on device)."""
        if self.x_scaler.scale != 0.0:
            xt = (xt - self.x_scaler.min_val) * (1.0 / self.x_scaler.scale)
        out = self.predict_normalized(xt).float()
        if self.residual_ridge is not None:
            if (self._ridge_t is None
                    or self._ridge_t.device != out.device):
                self._ridge_t = torch.from_numpy(np.asarray(
                    self.residual_ridge, dtype=np.float32)).to(out.device)
            N, T, P = xt.shape
            base = xt.reshape(-1, P) @ self._ridge_t[:P] + self._ridge_t[P]
            out = out + base.reshape(N, T, -1).unsqueeze(-1)
        # quantile regression can emit crossed quantiles (q95 < q50) early in
        # training; serving consumers (anomaly bands, demo plots) assume a
        # monotone triple, so sort the Q axis — a no-op once calibrated
        out, _ = torch.sort(out, dim=-1)
        if self.conformal is not None:
            # split-conformal band widening fitted at train time (CQR);
            # negative scores shrink the band — clamp at the median so the
            # served triple stays monotone (clamping a shrink only raises
            # coverage, the guarantee direction)
            if (self._conformal_t is None
                    or self._conformal_t.device != out.device):
                self._conformal_t = torch.from_numpy(np.asarray(
                    self.conformal, dtype=np.float32)).to(out.device)
            c = self._conformal_t
            out[..., 0] = torch.minimum(out[..., 0] - c, out[..., 1])
            out[..., -1] = torch.maximum(out[..., -1] + c, out[..., -2])
        out = out.cpu().numpy()                     # (N, T, M, Q)
        preds = {}
        for m, name in enumerate(self.metric_names):
            v = self.y_scalers[m].inverse_transform(out[:, :, m, :])
            if self.target_transform == "log1p":
                v = np.expm1(v)
            preds[name] = np.maximum(v, 1e-6)
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
