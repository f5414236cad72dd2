#!/usr/bin/env python3
"""Diagnose the accuracy-probe NaN under hipGraph-captured training.

Runs the probe app for a few epochs in four variants (graph on/off x
dropout 0/0.1) and prints per-epoch train losses + final DEEPR median.
"""
import argparse
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import numpy as np
import torch

from deeprest_amd.data.synthetic import SyntheticApp, SyntheticAppConfig
from deeprest_amd.engine.config import DataConfig, EngineConfig, TrainConfig
from deeprest_amd.engine.trainer import Trainer
from deeprest_amd.models.net import DeepRestNetConfig


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--epochs", type=int, default=10)
    args = ap.parse_args()
    device = torch.device("cuda" if torch.cuda.is_available() else "cpu")
    app = SyntheticApp(SyntheticAppConfig(
        n_apis=13, n_components=12, windows_per_day=240, n_days=8,
        resource_noise=0.03, seed=77))
    data = app.generate_featurized()
    for graph in (False, True):
        for dropout in (0.0, 0.1):
            cfg = EngineConfig(
                data=DataConfig(step_size=60, split=0.40),
                train=TrainConfig(epochs=args.epochs, batch_size=32, lr=1e-3,
                                  eval_cycles=9, baseline_epochs=2,
                                  run_baselines=False, log_every=0,
                                  eval_every=args.epochs, graph_step=graph),
                model=DeepRestNetConfig(dropout=dropout),
            )
            tr = Trainer(data, cfg, device=device)
            res = tr.train()
            deepr = [t["deepr"]["median"] for t in res.error_tables.values()]
            print(f"graph={graph} dropout={dropout} captured={tr.step.graphed} "
                  f"losses={[round(x, 4) for x in res.train_losses]} "
                  f"deepr_med={np.nanmean(deepr):.4f} "
                  f"nan_frac={np.mean(np.isnan(deepr)):.2f}", flush=True)


if __name__ == "__main__":
    main()
