#!/usr/bin/env python3
"""Unseen-traffic scenario evaluation at reference scale.

Trains once on normal diurnal traffic (reference config: 50 epochs, batch
32, window 60) and evaluates all four estimators on every unseen scenario
(3x scale, flat shape, unseen compositions) — the reference's headline
claim is accuracy on exactly these axes (README.md:3: ">90% even for unseen
traffic"). Prints the per-scenario error table and saves results.pkl.

  python tools/scenario_run.py [--epochs 50] [--apis 13] [--out results.pkl]
"""
import argparse
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import numpy as np
import torch

from deeprest_amd.data.synthetic import SyntheticApp, SyntheticAppConfig
from deeprest_amd.engine.config import DataConfig, EngineConfig, TrainConfig
from deeprest_amd.engine.experiment import (run_scenario_suite,
                                            scenario_error_tables,
                                            scenario_error_tables_by_resource)
from deeprest_amd.models.net import DeepRestNetConfig


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--epochs", type=int, default=50)
    ap.add_argument("--apis", type=int, default=13)
    ap.add_argument("--components", type=int, default=12)
    ap.add_argument("--days", type=int, default=8)
    ap.add_argument("--windows-per-day", type=int, default=240)
    ap.add_argument("--seed", type=int, default=77)
    ap.add_argument("--out", default="results_scenarios.pkl")
    ap.add_argument("--resources", type=int, choices=[3, 5], default=5,
                    help="3 = round-1 default, 5 = the reference's full set")
    ap.add_argument("--log-targets", action="store_true",
                    help="log1p target transform (unseen-scale extrapolation)")
    ap.add_argument("--no-residual", dest="residual", action="store_false",
                    help="disable the trace-ridge residual head (the "
                         "default head; carries unseen-scale extrapolation)")
    ap.add_argument("--no-conformal", action="store_true",
                    help="skip CQR band calibration")
    ap.set_defaults(residual=True)
    args = ap.parse_args()

    from deeprest_amd.data.synthetic import ALL_RESOURCES, DEFAULT_RESOURCES

    app = SyntheticApp(SyntheticAppConfig(
        n_apis=args.apis, n_components=args.components,
        windows_per_day=args.windows_per_day, n_days=args.days,
        resources=ALL_RESOURCES if args.resources == 5 else DEFAULT_RESOURCES,
        resource_noise=0.03, seed=args.seed))

    cfg = EngineConfig(
        data=DataConfig(step_size=60, split=0.40,
                        target_transform="log1p" if args.log_targets else "none"),
        train=TrainConfig(epochs=args.epochs, batch_size=32, lr=1e-3,
                          run_baselines=True, log_every=0,
                          residual_base="trace-ridge" if args.residual
                          else "none",
                          conformal=0.0 if args.no_conformal else 0.9),
        model=DeepRestNetConfig(dropout=0.1),
    )
    torch.manual_seed(0)
    dev = torch.device("cuda" if torch.cuda.is_available() else "cpu")
    store = run_scenario_suite(app, base_name="synthetic", config=cfg,
                               device=dev)
    store.save(args.out)

    def show(per_est, indent=""):
        for est in ("bl-resrc", "bl-api", "bl-trace", "ours"):
            t = per_est.get(est)
            if t is None:
                continue
            print(f"{indent}   {est:>9} => Median: {t['median']:.4f} | "
                  f"95-th: {t['p95']:.4f} | 99-th: {t['p99']:.4f} | "
                  f"Max: {t['max']:.4f}")

    tables = scenario_error_tables(store)
    by_res = scenario_error_tables_by_resource(store)
    for exp, per_est in tables.items():
        print(f"===== {exp} (all metrics aggregated) =====")
        show(per_est)
        for res, res_est in by_res[exp].items():
            print(f"  --- {res} ---")
            show(res_est, "  ")
    cov = getattr(store, "scenario_coverage", {})
    if cov:
        print("===== band coverage under scenario shift "
              "(nominal 0.90; non-monotone resources) =====")
        for exp, per_res in cov.items():
            parts = [f"{r}: {np.mean(v):.3f}" for r, v in per_res.items()]
            print(f"  {exp}: " + " | ".join(parts))
    print(f"saved {args.out}")


if __name__ == "__main__":
    main()
