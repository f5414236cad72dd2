#!/usr/bin/env python3
"""4096-endpoint training-step probe (BASELINE config 5 / VERDICT item 4).

Measures the full training step (forward + pinball + backward + fused
Adam) of the 4096-endpoint model at a ladder of window batches, with
synthetic-shaped random device tensors (the real featurized dataset at
this scale is ~50 GB of host transients; the step cost depends only on
shapes — stated openly here).  Reports windows/s, peak device memory and
free HBM per batch, and extrapolates the largest batch that fits in the
288 GB — WITHOUT running into OOM (memory ladder is bounded; each rung
only runs if predicted peak < 80% of free HBM).

  python tools/probe_4096.py [--batches 64,256,512] [--steps 5]
"""
import argparse
import json
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import time

import torch

from deeprest_amd.data.synthetic import SyntheticApp, SyntheticAppConfig
from deeprest_amd.engine.step import TrainStep
from deeprest_amd.models.net import DeepRestNet, DeepRestNetConfig, build_model_spec
from deeprest_amd.ops.adam import FusedAdam


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--endpoints", type=int, default=4096)
    ap.add_argument("--components", type=int, default=63)
    ap.add_argument("--batches", default="64,256,512")
    ap.add_argument("--steps", type=int, default=5)
    ap.add_argument("--warmup", type=int, default=2)
    ap.add_argument("--seq-len", type=int, default=60)
    args = ap.parse_args()
    dev = torch.device("cuda")

    # small windows_per_day: we only need the app's STRUCTURE (feature
    # space / spec); step tensors are synthetic-shaped randoms on device
    t0 = time.perf_counter()
    app = SyntheticApp(SyntheticAppConfig(
        n_apis=args.endpoints, n_components=args.components,
        windows_per_day=16, n_days=1, seed=7))
    data = app.generate_featurized()
    spec = build_model_spec(data)
    P, T, M = spec.num_paths, args.seq_len, spec.num_metrics
    print(f"setup {time.perf_counter() - t0:.1f}s paths={P} metrics={M}",
          flush=True)

    torch.manual_seed(0)
    model = DeepRestNet(spec, DeepRestNetConfig(dropout=0.0)).to(dev)
    opt = FusedAdam(model.parameters(), lr=1e-3, capturable=True)
    step = TrainStep(model, opt, autocast_dtype=torch.bfloat16)

    results = []
    prev = None
    for B in (int(b) for b in args.batches.split(",")):
        free, total = torch.cuda.mem_get_info()
        if prev is not None:
            # linear-in-B activation model from the previous rung
            pred_peak = prev["peak_gb"] / prev["batch"] * B * 1.15
            if pred_peak > 0.80 * free / 2**30 + prev["peak_gb"]:
                print(f"B={B}: predicted peak {pred_peak:.0f} GB exceeds "
                      f"80% of free HBM — skipping (bounded ladder)",
                      flush=True)
                continue
        X = torch.rand(B, T, P, device=dev, dtype=torch.bfloat16)
        y = torch.rand(B, T, M, device=dev)
        torch.cuda.reset_peak_memory_stats()
        for _ in range(args.warmup):
            step(X, y)
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(args.steps):
            step(X, y)
        torch.cuda.synchronize()
        dt = time.perf_counter() - t0
        peak = torch.cuda.max_memory_allocated() / 2**30
        free, total = torch.cuda.mem_get_info()
        r = {"batch": B, "windows_per_s": round(B * args.steps / dt, 1),
             "ms_per_step": round(dt / args.steps * 1000, 1),
             "peak_gb": round(peak, 1),
             "free_gb": round(free / 2**30, 1),
             "total_gb": round(total / 2**30, 1)}
        print(json.dumps(r), flush=True)
        results.append(r)
        prev = r
        del X, y
        torch.cuda.empty_cache()

    if len(results) >= 2:
        # per-window activation cost from the ladder; extrapolate max batch
        a, b = results[-2], results[-1]
        per_win = (b["peak_gb"] - a["peak_gb"]) / (b["batch"] - a["batch"])
        fixed = a["peak_gb"] - per_win * a["batch"]
        budget = results[-1]["total_gb"] * 0.92
        max_b = int((budget - fixed) / per_win)
        print(json.dumps({
            "per_window_gb": round(per_win, 4),
            "fixed_gb": round(fixed, 1),
            "hbm_budget_gb": round(budget, 1),
            "extrapolated_max_batch": max_b,
        }))


if __name__ == "__main__":
    main()
