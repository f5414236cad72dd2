"""Training loop with the three-estimator comparison harness.

Mirrors the reference driver's behavior (reference:
resource-estimation/estimate.py:21-123): run both baselines up front, train
the estimator with quantile loss + Adam, and per evaluation report the
Median/95th/99th/Max absolute-error table for RESRC, COMP and DEEPR on the
non-overlapping held-out windows, denormalizing the model's median quantile.

Additions over the reference: checkpoint/resume, bf16 autocast on GPU,
fused-Adam optimizer, optional data parallelism (one process per GPU over
RCCL; gradients leave through the DistContext hook so the loop itself stays
backend-agnostic and testable with gloo on CPU).
"""

from __future__ import annotations

import time
from dataclasses import dataclass, field
from typing import Dict, List, Optional

import numpy as np
import torch

from ..data.featurize import FeaturizedData
from ..models.baselines import ComponentAwareBaseline
from ..models.net import DeepRestNet, build_model_spec
from ..ops.adam import FusedAdam
from ..utils.errors import (error_percentiles, format_error_table,
                            quantile_coverage)
from .checkpoint import load_checkpoint, save_checkpoint
from .config import EngineConfig
from .dataset import EstimationDataset
from .step import TrainStep


@dataclass
class TrainResult:
    train_losses: List[float] = field(default_factory=list)
    test_losses: List[float] = field(default_factory=list)
    error_tables: Dict[str, Dict[str, Dict[str, float]]] = field(default_factory=dict)
    # per-metric calibration of the outer quantile band on the eval windows
    # (nominal 0.90 for (.05, .95)); see utils.errors.quantile_coverage
    coverage: Dict[str, Dict[str, float]] = field(default_factory=dict)
    samples_per_sec: float = 0.0

    def summary(self) -> str:
        lines = []
        for name, per_est in self.error_tables.items():
            lines.append(format_error_table(name, per_est))
        return "\n".join(lines)


class Trainer:
    def __init__(
        self,
        data: FeaturizedData,
        config: Optional[EngineConfig] = None,
        device: Optional[torch.device] = None,
        dist_ctx=None,
        model: Optional[DeepRestNet] = None,
    ) -> None:
        self.cfg = config or EngineConfig()
        self.device = device or torch.device(
            "cuda" if torch.cuda.is_available() else "cpu"
        )
        self.dist = dist_ctx
        self.rank = dist_ctx.rank if dist_ctx is not None else 0
        self.world_size = dist_ctx.world_size if dist_ctx is not None else 1

        self.dataset = EstimationDataset(
            data, step_size=self.cfg.data.step_size,
            split_fraction=self.cfg.data.split,
            target_transform=self.cfg.data.target_transform,
        )
        if model is None:
            spec = build_model_spec(data)
            model = DeepRestNet(spec, self.cfg.model)
        self.model = model.to(self.device)
        if dist_ctx is not None and dist_ctx.world_size > 1:
            # replicas must start identical: model construction is unseeded,
            # so sync rank-0 weights before the first averaged-gradient step
            dist_ctx.broadcast_parameters(self.model)
        # graph_step wants the capturable optimizer (device-side step counter);
        # capturable mode is also correct for plain eager stepping
        self._use_graph = bool(
            self.cfg.train.graph_step
            and self.device.type == "cuda"
            and (dist_ctx is None or dist_ctx.world_size == 1)
        )
        self.optimizer = FusedAdam(self.model.parameters(), lr=self.cfg.train.lr,
                                   capturable=self._use_graph)
        self.feature_space_state = (
            data.feature_space.state_dict() if data.feature_space is not None else None
        )
        self.start_epoch = 0
        self._baseline_preds: Optional[Dict[str, np.ndarray]] = None
        # ridge-residual base (train.residual_base="trace-ridge"): the net
        # learns y_norm - ridge(x_norm); the ridge carries linear
        # extrapolation to unseen traffic scale (the bounded net saturates
        # outside its fitted min-max — profiles/r02_unseen_traffic.md)
        self._ridge_w: Optional[np.ndarray] = None       # (P+1, M) f64
        self._ridge_norm: Optional[np.ndarray] = None    # (N, T, M) f32
        self._conformal: Optional[np.ndarray] = None     # (M,) band widening
        if self.cfg.train.residual_base == "trace-ridge":
            self._fit_residual_base()
        elif self.cfg.train.residual_base != "none":
            raise ValueError(
                f"unknown residual_base {self.cfg.train.residual_base!r}")
        self.autocast_dtype = (
            torch.bfloat16 if self.cfg.train.dtype == "bf16" else torch.float32
        )
        # THE training step — shared verbatim with bench.py (engine/step.py)
        self.step = TrainStep(
            self.model, self.optimizer, dist_ctx=self.dist,
            autocast_dtype=self.autocast_dtype
            if self.device.type == "cuda" else None,
        )

    # -------------------------------------------------------- residual base
    def _fit_residual_base(self) -> None:
        ds = self.dataset
        X = ds.X.numpy()
        N, T, P = X.shape
        if P > 20000:
            raise ValueError(
                "trace-ridge residual base solves (P+1)^2 normal equations; "
                f"P={P} is past the closed-form range (use <= ~20k paths)")
        M = ds.y.shape[-1]
        Xf = X[: ds.split].reshape(-1, P).astype(np.float64)
        Xf = np.concatenate([Xf, np.ones((len(Xf), 1))], axis=1)
        yf = ds.y[: ds.split].numpy().reshape(-1, M).astype(np.float64)
        A = Xf.T @ Xf + 1e-3 * np.eye(P + 1)
        self._ridge_w = np.linalg.solve(A, Xf.T @ yf)          # (P+1, M)
        self._ridge_norm = self.ridge_apply(X).astype(np.float32)

    def ridge_apply(self, x_norm: np.ndarray) -> Optional[np.ndarray]:
        """Normalized windows (N, T, P) -> normalized-space ridge predictions
        (N, T, M); None when no residual base is fitted."""
        if self._ridge_w is None:
            return None
        N, T, P = x_norm.shape
        flat = (x_norm.reshape(-1, P) @ self._ridge_w[:P]
                + self._ridge_w[P])
        return flat.reshape(N, T, -1)

    # ------------------------------------------------------------- baselines
    def run_baselines(self) -> Dict[str, np.ndarray]:
        """Fit both baselines per metric; predictions for ALL test windows."""
        if self._baseline_preds is not None:
            return self._baseline_preds
        ds = self.dataset
        t_cfg = self.cfg.train
        # RESRC: every metric's history-MLP fit in ONE batched run (on the
        # GPU when present; 39 sequential CPU fits took 74 s)
        from ..models.baselines import ResourceAwareBatchBaseline

        y_stack = np.ascontiguousarray(np.moveaxis(ds.y_raw, -1, 0))  # (M, N, T)
        resrc_all = ResourceAwareBatchBaseline(
            split=ds.split, window=ds.step_size,
            epochs=t_cfg.baseline_epochs, seed=t_cfg.seed,
            device=self.device if self.device.type == "cuda" else None,
        ).fit_and_estimate(y_stack)                         # (M, N_test, T)
        comp_list = []
        for idx, name in enumerate(ds.metric_names):
            yw = ds.y_raw[:, :, idx]
            comp_name = self.model.spec.components[self.model.spec.comp_of[idx]]
            comp = ComponentAwareBaseline(
                component=comp_name, invocations=ds.data.invocations,
                window=ds.step_size, split=ds.split,
            ).fit_and_estimate(yw)
            comp_list.append(comp[:, :, None])
        self._baseline_preds = {
            "resrc": np.moveaxis(resrc_all, 0, -1),         # (N_test, T, M)
            "comp": np.concatenate(comp_list, axis=-1),
        }
        return self._baseline_preds

    # ------------------------------------------------------------------ train
    def train(self) -> TrainResult:
        cfg = self.cfg.train
        ds = self.dataset
        result = TrainResult()

        baselines = self.run_baselines() if cfg.run_baselines and self.rank == 0 else None

        if cfg.resume and cfg.checkpoint_path:
            try:
                self.load(cfg.checkpoint_path)
            except FileNotFoundError:
                pass

        X_train = ds.X_train.to(self.device)
        y_train = ds.y_train
        if self._ridge_norm is not None:
            # train on the residual: quantiles are shift-equivariant, so
            # residual quantiles + ridge = target quantiles
            y_train = y_train - torch.from_numpy(self._ridge_norm[: ds.split])
        y_train = y_train.to(self.device)
        n = X_train.shape[0]
        gen = torch.Generator().manual_seed(cfg.seed)
        # resume determinism: replay the skipped epochs' permutation draws so
        # a resumed run sees the same batch order as an uninterrupted one
        for _ in range(self.start_epoch):
            torch.randperm(n, generator=gen)

        total_samples = 0
        t_start = time.perf_counter()
        for epoch in range(self.start_epoch, cfg.epochs):
            if cfg.lr_schedule == "cosine":
                # decay to 5% of the base lr; graph-safe — FusedAdam.set_lr
                # refreshes the device scalar captured replays read
                frac = epoch / max(cfg.epochs - 1, 1)
                lr_now = cfg.lr * (0.05 + 0.95 * 0.5 * (1.0 + np.cos(np.pi * frac)))
                if hasattr(self.optimizer, "set_lr"):
                    self.optimizer.set_lr(lr_now)
                else:
                    for group in self.optimizer.param_groups:
                        group["lr"] = lr_now
            self.model.train()
            perm = torch.randperm(n, generator=gen)
            # data-parallel shard: equal-length strided slices — truncate to a
            # multiple of world_size first, else ranks can issue different
            # numbers of all_reduce calls per epoch (collectives pairing across
            # batches = silently corrupted averaging, then a hang at the end)
            if self.world_size > 1:
                n_even = n - (n % self.world_size)
                perm = perm[:n_even][self.rank :: self.world_size]
            losses = []
            for s in range(0, len(perm), cfg.batch_size):
                idx = perm[s : s + cfg.batch_size].to(self.device)
                xb, yb = X_train[idx], y_train[idx]
                if (self._use_graph and not self.step.graphed
                        and xb.shape[0] == cfg.batch_size):
                    # capture once on the first full batch (warmup replays
                    # are real optimizer steps on that batch); tail batches
                    # and capture failure fall back to eager inside TrainStep
                    if not self.step.try_capture(xb, yb):
                        self._use_graph = False
                loss = self.step(xb, yb)
                # keep the loss on-device: a per-batch .item() is a host sync
                # that drains the GPU pipeline every step
                losses.append(loss.detach().clone())
                total_samples += xb.shape[0] * self.world_size
            result.train_losses.append(
                float(torch.stack(losses).mean().item()) if losses else float("nan")
            )

            do_eval = ((epoch + 1) % max(cfg.eval_every, 1) == 0
                       or epoch + 1 == cfg.epochs)
            if self.rank == 0 and do_eval:
                test_loss, tables = self.evaluate(baselines)
                result.test_losses.append(test_loss)
                result.error_tables = tables
                result.coverage = getattr(self, "last_coverage", {})
                if cfg.log_every and (epoch + 1) % cfg.log_every == 0:
                    print(
                        f"Epoch [{epoch + 1}/{cfg.epochs}], "
                        f"Train Loss: {result.train_losses[-1]:.6f}, "
                        f"Test Loss: {test_loss:.6f}"
                    )
                    for name, per_est in tables.items():
                        print(format_error_table(name, per_est))
                if cfg.checkpoint_path:
                    self.save(cfg.checkpoint_path, epoch + 1)

        elapsed = time.perf_counter() - t_start
        result.samples_per_sec = total_samples / max(elapsed, 1e-9)

        if self.rank == 0 and cfg.conformal > 0.0:
            self.conformalize(cfg.conformal)
            # re-evaluate so the reported coverage reflects the widened band
            test_loss, tables = self.evaluate(baselines)
            result.test_losses.append(test_loss)
            result.error_tables = tables
            result.coverage = getattr(self, "last_coverage", {})
            if cfg.checkpoint_path:
                # the widening travels with the checkpoint (Predictor reads it)
                self.save(cfg.checkpoint_path, cfg.epochs)
        return result

    # ----------------------------------------------------------- conformal
    @torch.no_grad()
    def conformalize(self, target: float = 0.9, max_windows: int = 512) -> None:
        """Split-conformal band widening (conformalized quantile regression):
        on CALIBRATION windows — test-split windows disjoint from the
        non-overlapping eval set — compute per metric the conformity score
        s = max(q_lo - y, y - q_hi) in normalized space and widen the outer
        band by its ceil((n+1)(1-a))/n quantile, giving finite-sample
        coverage >= target under exchangeability.  The median is untouched,
        so the error tables are unchanged."""
        ds = self.dataset
        eval_set = set(ds.eval_window_indices(self.cfg.train.eval_cycles))
        n_test = ds.num_windows - ds.split
        cal_idx = [iv for iv in range(n_test) if iv not in eval_set]
        if not cal_idx:
            cal_idx = list(range(n_test))      # degenerate tiny datasets
        cal_idx = cal_idx[:max_windows]
        self.model.eval()
        xb = ds.X_test[cal_idx].to(self.device)
        yb = ds.y_test[cal_idx].cpu().numpy()            # normalized (K,T,M)
        with torch.autocast(
            device_type=self.device.type, dtype=self.autocast_dtype,
            enabled=(self.device.type == "cuda"),
        ):
            out = self.model(xb)
        out = out.float().cpu().numpy()
        rb = self.ridge_apply(ds.X_test[cal_idx].numpy())
        if rb is not None:
            out = out + rb[..., None]
        band = np.sort(out, axis=-1)                     # (K,T,M,Q)
        Q = band.shape[-1]
        alpha = 1.0 - target
        M = yb.shape[-1]
        offs = np.zeros(M, dtype=np.float32)
        for m in range(M):
            s = np.maximum(band[:, :, m, 0] - yb[:, :, m],
                           yb[:, :, m] - band[:, :, m, Q - 1]).ravel()
            n = len(s)
            q = min((np.ceil((n + 1) * (1.0 - alpha)) / n), 1.0)
            offs[m] = np.quantile(s, q)
        self._conformal = offs

    # ------------------------------------------------------------------- eval
    @torch.no_grad()
    def evaluate(self, baselines: Optional[Dict[str, np.ndarray]] = None):
        """Test loss + per-metric error tables on non-overlapping windows."""
        cfg = self.cfg.train
        ds = self.dataset
        self.model.eval()
        eval_idx = ds.eval_window_indices(cfg.eval_cycles)
        if not eval_idx:
            return float("nan"), {}
        xb = ds.X_test[eval_idx].to(self.device)
        yb = ds.y_test[eval_idx].to(self.device)
        with torch.autocast(
            device_type=self.device.type, dtype=self.autocast_dtype,
            enabled=(self.device.type == "cuda"),
        ):
            out = self.model(xb)
        out = out.float()
        if self._ridge_norm is not None:
            ridge = torch.from_numpy(
                self._ridge_norm[ds.split :][eval_idx]).to(self.device)
            out = out + ridge.unsqueeze(-1)
        test_loss = float(self.model.loss(out, yb).item())

        outputs = np.maximum(out.cpu().numpy(), 1e-6)      # (K, T, M, Q)
        labels = yb.cpu().numpy()
        tables: Dict[str, Dict[str, Dict[str, float]]] = {}
        coverage: Dict[str, Dict[str, float]] = {}
        Q = outputs.shape[-1]
        median_q = len(self.model.cfg.quantiles) // 2
        for m, name in enumerate(ds.metric_names):
            labels_d = ds.denormalize_metric(labels[:, :, m], m).ravel()
            pred_d = ds.denormalize_metric(outputs[:, :, m, median_q], m).ravel()
            per_est = {}
            if baselines is not None:
                for est_name, key in (("resrc", "resrc"), ("comp", "comp")):
                    bl = baselines[key][eval_idx][:, :, m].ravel()
                    per_est[est_name] = error_percentiles(np.abs(bl - labels_d))
            per_est["deepr"] = error_percentiles(np.abs(pred_d - labels_d))
            tables[name] = per_est
            # outer-band calibration (quantiles sorted: serving semantics;
            # conformal widening applied when fitted)
            band = np.sort(outputs[:, :, m, :], axis=-1)
            if self._conformal is not None:
                # negative scores shrink; clamp at the median (serving
                # semantics, predictor.py)
                band[:, :, 0] = np.minimum(
                    band[:, :, 0] - self._conformal[m], band[:, :, 1])
                band[:, :, -1] = np.maximum(
                    band[:, :, -1] + self._conformal[m], band[:, :, -2])
            coverage[name] = quantile_coverage(
                labels_d,
                ds.denormalize_metric(band[:, :, 0], m).ravel(),
                ds.denormalize_metric(band[:, :, Q - 1], m).ravel())
        self.last_coverage = coverage
        return test_loss, tables

    # ------------------------------------------------------------ checkpoints
    def save(self, path: str, epoch: int) -> None:
        save_checkpoint(
            path,
            self.model,
            optimizer=self.optimizer,
            scaler_state=self.dataset.scaler_state(),
            feature_space_state=self.feature_space_state,
            epoch=epoch,
            extra={"config": self.cfg.to_dict(),
                   "residual_ridge": self._ridge_w,
                   "conformal": self._conformal},
        )

    def load(self, path: str) -> None:
        state = load_checkpoint(path, map_location=self.device)
        self.model.load_state_dict(state["model"]["state_dict"])
        if state.get("optimizer"):
            self.optimizer.load_state_dict(state["optimizer"])
        self.start_epoch = int(state.get("epoch", 0))
