"""The two comparison baselines from the reference's validation harness.

These ARE part of the framework contract: every training session runs them and
reports the three-way RESRC/COMP/DEEPR error table
(reference: resource-estimation/estimate.py:31-39,112-122).

- ResourceAwareBaseline: API-blind history autoregressor — an MLP that maps
  the utilization window `offset` steps back to the next window; at test time
  a single predicted window is repeated (reference: baselines.py:34-77).
- ComponentAwareBaseline: API-aware only at component granularity — a min-max
  linear scaling from the component's invocation counts to the metric
  (reference: baselines.py:80-110).
"""

from __future__ import annotations

from typing import Dict, Optional

import numpy as np
import torch
import torch.nn as nn

from ..data.windows import minmax_fit, minmax_apply


class ResourceAwareBaseline(nn.Module):
    """History-only MLP autoregressor over utilization windows."""

    def __init__(
        self,
        split: int,
        window: int,
        offset: Optional[int] = None,
        hidden: int = 128,
        epochs: int = 100,
        lr: float = 1e-3,
        batch_size: int = 32,
        device: Optional[torch.device] = None,
        seed: int = 0,
    ) -> None:
        super().__init__()
        self.split = split
        self.window = window
        self.offset = offset if offset is not None else window - 1
        self.epochs = epochs
        self.lr = lr
        self.batch_size = batch_size
        gen = torch.Generator().manual_seed(seed)
        self.net = nn.Sequential(
            nn.Linear(window, hidden), nn.ReLU(), nn.Linear(hidden, window)
        )
        # deterministic init independent of global RNG state
        with torch.no_grad():
            for m in self.net:
                if isinstance(m, nn.Linear):
                    bound = 1.0 / np.sqrt(m.in_features)
                    m.weight.uniform_(-bound, bound, generator=gen)
                    m.bias.uniform_(-bound, bound, generator=gen)
        self.device = device or torch.device("cpu")
        self.to(self.device)

    def fit_and_estimate(self, y_windows: np.ndarray) -> np.ndarray:
        """y_windows: (N, window) one metric's windowed utilization.

        Returns (N - split, window) predictions for the test windows —
        a single next-window estimate repeated, as the reference does
        (baselines.py:69-77).
        """
        y = np.asarray(y_windows, dtype=np.float64)
        lo, hi = minmax_fit(y, self.split)
        yn = minmax_apply(y, lo, hi)
        scale = hi - lo

        # (input window `offset` steps back) -> (current window)
        xs, ys = [], []
        for i in range(self.offset, len(yn)):
            xs.append(yn[i - self.offset])
            ys.append(yn[i])
        X = np.asarray(xs, dtype=np.float32)
        Y = np.asarray(ys, dtype=np.float32)
        train_n = self.split - self.offset
        if train_n <= 0:
            raise ValueError("split must exceed offset for the history baseline")

        Xt = torch.from_numpy(X[:train_n]).to(self.device)
        Yt = torch.from_numpy(Y[:train_n]).to(self.device)
        opt = torch.optim.Adam(self.parameters(), lr=self.lr)
        loss_fn = nn.MSELoss()
        n = Xt.shape[0]
        for _ in range(self.epochs):
            perm = torch.randperm(n)
            for s in range(0, n, self.batch_size):
                idx = perm[s : s + self.batch_size]
                opt.zero_grad()
                loss = loss_fn(self.net(Xt[idx]), Yt[idx])
                loss.backward()
                opt.step()

        with torch.no_grad():
            probe = torch.from_numpy(X[[train_n - self.offset]]).to(self.device) \
                if train_n - self.offset >= 0 else Xt[[-1]]
            pred = self.net(probe).cpu().numpy().squeeze(0)
        pred = np.maximum(pred * scale + lo, 1e-6)
        n_test = len(y) - self.split
        return np.tile(pred, (n_test, 1))


class ResourceAwareBatchBaseline:
    """All metrics' history MLPs trained together as one batched model.

    Semantics match running ResourceAwareBaseline per metric (reference:
    baselines.py:34-77): independent per-metric weights (stacked along a
    leading M axis, driven by bmm), the same shared init (the sequential
    harness seeds every metric's MLP identically), per-metric min-max
    normalization, and a per-metric-summed MSE so each metric's gradients
    are exactly what its independent fit would see (Adam is elementwise).
    ~M x fewer op launches: 39 sequential fits took 74 s on CPU; this runs
    them in one loop (and on the GPU when one is present).
    """

    def __init__(
        self,
        split: int,
        window: int,
        offset: Optional[int] = None,
        hidden: int = 128,
        epochs: int = 100,
        lr: float = 1e-3,
        batch_size: int = 32,
        device: Optional[torch.device] = None,
        seed: int = 0,
    ) -> None:
        self.split = split
        self.window = window
        self.offset = offset if offset is not None else window - 1
        self.hidden = hidden
        self.epochs = epochs
        self.lr = lr
        self.batch_size = batch_size
        self.seed = seed
        self.device = device or torch.device("cpu")

    def fit_and_estimate(self, y_stack: np.ndarray) -> np.ndarray:
        """y_stack: (M, N, window) -> (M, N - split, window) predictions."""
        y = np.asarray(y_stack, dtype=np.float64)
        M, N, W = y.shape
        train_n = self.split - self.offset
        if train_n <= 0:
            raise ValueError("split must exceed offset for the history baseline")

        # per-metric min-max on the train split (minmax_fit/apply semantics,
        # incl. the rng==0 "leave unchanged" edge case)
        lo = y[:, : self.split].reshape(M, -1).min(axis=1)
        hi = y[:, : self.split].reshape(M, -1).max(axis=1)
        rng = hi - lo
        div = np.where(rng != 0.0, rng, 1.0)
        sub = np.where(rng != 0.0, lo, 0.0)
        yn = (y - sub[:, None, None]) / div[:, None, None]

        X = np.ascontiguousarray(yn[:, :train_n], dtype=np.float32)
        # target = window `offset` steps ahead of the input window
        Y = np.ascontiguousarray(
            yn[:, self.offset : self.offset + train_n], dtype=np.float32)

        dev = self.device
        Xt = torch.from_numpy(X).to(dev)           # (M, n, W)
        Yt = torch.from_numpy(Y).to(dev)
        gen = torch.Generator().manual_seed(self.seed)
        bound1 = 1.0 / np.sqrt(W)
        bound2 = 1.0 / np.sqrt(self.hidden)
        # one init shared by all metrics — the sequential harness seeds every
        # metric's MLP with the same generator, so their inits are identical.
        # Draw in nn.Linear's (out, in) layout so the generator stream maps
        # to the same elements, then transpose for bmm.
        w1 = torch.empty(self.hidden, W).uniform_(
            -bound1, bound1, generator=gen).t().contiguous()
        b1 = torch.empty(self.hidden).uniform_(-bound1, bound1, generator=gen)
        w2 = torch.empty(W, self.hidden).uniform_(
            -bound2, bound2, generator=gen).t().contiguous()
        b2 = torch.empty(W).uniform_(-bound2, bound2, generator=gen)
        w1 = w1[None].repeat(M, 1, 1).to(dev).requires_grad_(True)
        b1 = b1[None, None].repeat(M, 1, 1).to(dev).requires_grad_(True)
        w2 = w2[None].repeat(M, 1, 1).to(dev).requires_grad_(True)
        b2 = b2[None, None].repeat(M, 1, 1).to(dev).requires_grad_(True)

        opt = torch.optim.Adam([w1, b1, w2, b2], lr=self.lr)
        n = Xt.shape[1]
        perm_gen = torch.Generator().manual_seed(self.seed + 1)
        for _ in range(self.epochs):
            perm = torch.randperm(n, generator=perm_gen)
            for s in range(0, n, self.batch_size):
                idx = perm[s : s + self.batch_size].to(dev)
                xb = Xt[:, idx]                     # (M, B, W)
                yb = Yt[:, idx]
                h = torch.relu(torch.bmm(xb, w1) + b1)
                pred = torch.bmm(h, w2) + b2
                # sum of per-metric means: each metric's grads match its
                # independent fit exactly
                loss = ((pred - yb) ** 2).mean(dim=(1, 2)).sum()
                opt.zero_grad()
                loss.backward()
                opt.step()

        with torch.no_grad():
            pi = train_n - self.offset
            probe = Xt[:, [pi if pi >= 0 else -1]]  # (M, 1, W)
            h = torch.relu(torch.bmm(probe, w1) + b1)
            pred = (torch.bmm(h, w2) + b2).squeeze(1).cpu().numpy()  # (M, W)
        pred = np.maximum(pred * rng[:, None] + lo[:, None], 1e-6)
        n_test = N - self.split
        return np.tile(pred[:, None, :], (1, n_test, 1))


class TraceAwareBaseline:
    """Ridge regression from call-path traffic features to the metric.

    The third comparison estimator the demo UI expects ('bl-trace',
    reference: web-demo/dataloader.py:112-125): API-aware at call-path
    granularity but linear — no temporal modeling.  Fit on the train split
    by closed-form ridge; predictions are per-window.
    """

    def __init__(self, split: int, ridge: float = 1e-3) -> None:
        self.split = split
        self.ridge = ridge
        self._w: Optional[np.ndarray] = None

    def fit(self, X_windows: np.ndarray, y_windows: np.ndarray) -> "TraceAwareBaseline":
        X = np.asarray(X_windows, dtype=np.float64)
        y = np.asarray(y_windows, dtype=np.float64)
        P = X.shape[-1]
        Xf = X[: self.split].reshape(-1, P)
        yf = y[: self.split].reshape(-1)
        # bias column + ridge-regularized normal equations
        Xf = np.concatenate([Xf, np.ones((len(Xf), 1))], axis=1)
        A = Xf.T @ Xf + self.ridge * np.eye(P + 1)
        self._w = np.linalg.solve(A, Xf.T @ yf)
        return self

    def estimate(self, X_windows: np.ndarray) -> np.ndarray:
        """(N, T, P) windows (any traffic, incl. unseen scenarios) -> (N, T)."""
        assert self._w is not None, "fit first"
        X = np.asarray(X_windows, dtype=np.float64)
        N, T, P = X.shape
        Xt = X.reshape(-1, P)
        Xt = np.concatenate([Xt, np.ones((len(Xt), 1))], axis=1)
        pred = (Xt @ self._w).reshape(-1, T)
        return np.maximum(pred, 1e-6)

    def fit_and_estimate(self, X_windows: np.ndarray, y_windows: np.ndarray) -> np.ndarray:
        """X_windows: (N, T, P); y_windows: (N, T). Returns (N - split, T)."""
        self.fit(X_windows, y_windows)
        return self.estimate(np.asarray(X_windows)[self.split :])


class ComponentAwareBaseline:
    """Min-max linear scaling from component invocation counts to the metric."""

    def __init__(self, component: str, invocations: Dict[str, np.ndarray],
                 window: int, split: int) -> None:
        self.window = window
        self.split = split
        inv = invocations.get(component)
        self.invocation = np.asarray(
            inv if inv is not None else invocations["general"], dtype=np.float64
        )

    def fit(self, y_windows: np.ndarray) -> "ComponentAwareBaseline":
        """Fit the min-max scaling map (w1..w4) on the train portion."""
        y = np.asarray(y_windows, dtype=np.float64)
        # reconstruct the flat series from stride-1 windows
        ts = np.concatenate([y[:-1, 0], y[-1]]) if len(y) > 1 else y[0]
        split_flat = self.split + self.window - 1

        inv_train = self.invocation[:split_flat]
        met_train = ts[:split_flat]
        self.w1 = float(np.min(inv_train))
        self.w3 = float(np.max(inv_train) - self.w1)
        self.w4 = float(np.min(met_train))
        self.w2 = float(np.max(met_train) - self.w4)
        self._n_flat = len(ts)
        return self

    def estimate_series(self, invocation: np.ndarray, n_flat: int) -> np.ndarray:
        """Apply the fitted map to ANY invocation series (e.g. an unseen
        scenario's); returns stride-1 windows over its first n_flat steps."""
        inv = np.asarray(invocation, dtype=np.float64)
        if inv.sum() > 0 and self.w3 > 0:
            ts_hat = (inv - self.w1) * self.w2 / self.w3 + self.w4
        else:
            ts_hat = inv.astype(np.float64)
        ts_hat = np.maximum(ts_hat, 1e-6)
        return np.stack(
            [ts_hat[i - self.window : i] for i in range(self.window, n_flat + 1)]
        )

    def fit_and_estimate(self, y_windows: np.ndarray) -> np.ndarray:
        """y_windows: (N, window). Returns (N - split, window) estimates."""
        self.fit(y_windows)
        windows = self.estimate_series(self.invocation, self._n_flat)
        return windows[self.split :]
