#!/usr/bin/env python3
"""Isolate the sporadic NaN: eager-vs-graph x capturable-vs-plain Adam.

Variants:
  plain    — eager steps, FusedAdam(capturable=False)   (round-1 behavior)
  cap      — eager steps, FusedAdam(capturable=True)    (device step counter)
  graph    — hipGraph-captured step (implies capturable)

On the first non-finite epoch loss, dumps which tensors (param / grad /
exp_avg / exp_avg_sq) contain non-finite values.
"""
import argparse
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import torch

from deeprest_amd.data.synthetic import SyntheticApp, SyntheticAppConfig
from deeprest_amd.engine.dataset import EstimationDataset
from deeprest_amd.engine.step import TrainStep
from deeprest_amd.models.net import DeepRestNet, DeepRestNetConfig, build_model_spec
from deeprest_amd.ops.adam import FusedAdam


def run_variant(variant, data, device, epochs, seed):
    torch.manual_seed(seed)
    ds = EstimationDataset(data, step_size=60, split_fraction=0.40)
    spec = build_model_spec(data)
    model = DeepRestNet(spec, DeepRestNetConfig(dropout=0.0)).to(device)
    capturable = variant in ("cap", "graph")
    opt = FusedAdam(model.parameters(), lr=1e-3, capturable=capturable)
    step = TrainStep(model, opt, autocast_dtype=torch.bfloat16)
    X = ds.X_train.to(device)
    y = ds.y_train.to(device)
    n = X.shape[0]
    bs = 32
    gen = torch.Generator().manual_seed(seed)
    captured = False
    for epoch in range(epochs):
        perm = torch.randperm(n, generator=gen)
        losses = []
        for s in range(0, n, bs):
            idx = perm[s : s + bs].to(device)
            xb, yb = X[idx], y[idx]
            if variant == "graph" and not captured and xb.shape[0] == bs:
                captured = step.try_capture(xb, yb)
            losses.append(step(xb, yb).detach().clone())
        mean = torch.stack(losses).mean().item()
        if not (mean == mean and abs(mean) < 1e9):
            print(f"  !! non-finite at epoch {epoch}: {mean}")
            for name, p in model.named_parameters():
                st = opt.state.get(p, {})
                bad = []
                if not torch.isfinite(p).all():
                    bad.append("param")
                if p.grad is not None and not torch.isfinite(p.grad).all():
                    bad.append("grad")
                for k in ("exp_avg", "exp_avg_sq"):
                    if k in st and not torch.isfinite(st[k]).all():
                        bad.append(k)
                if bad:
                    print(f"    {name}: {','.join(bad)}")
            return epoch, mean
        if epoch % 5 == 0:
            print(f"  epoch {epoch}: {mean:.4f}", flush=True)
    return None, mean


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--epochs", type=int, default=30)
    ap.add_argument("--variants", default="plain,cap,graph")
    ap.add_argument("--seeds", default="0,1")
    args = ap.parse_args()
    device = torch.device("cuda" if torch.cuda.is_available() else "cpu")
    app = SyntheticApp(SyntheticAppConfig(
        n_apis=13, n_components=12, windows_per_day=240, n_days=8,
        resource_noise=0.03, seed=77))
    data = app.generate_featurized()
    for variant in args.variants.split(","):
        for seed in (int(s) for s in args.seeds.split(",")):
            print(f"variant={variant} seed={seed}", flush=True)
            bad_epoch, final = run_variant(variant, data, device,
                                           args.epochs, seed)
            print(f"variant={variant} seed={seed} -> "
                  f"{'NaN@' + str(bad_epoch) if bad_epoch is not None else f'ok final={final:.4f}'}",
                  flush=True)


if __name__ == "__main__":
    main()
