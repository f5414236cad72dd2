#!/usr/bin/env python3
"""End-to-end walkthrough on a synthetic 8-endpoint application.

Covers the full capability surface in one script: generate -> featurize ->
train with the three-estimator harness -> what-if synthesis -> anomaly
check -> results.pkl. CPU-friendly (small config); pass --gpu to run the
estimation engine on an MI355X with the HIP kernels.
"""
import argparse
import os
import sys
sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import numpy as np
import torch

from deeprest_amd.data.featurize import Featurizer
from deeprest_amd.data.synthesizer import TraceSynthesizer
from deeprest_amd.data.synthetic import SyntheticApp, SyntheticAppConfig
from deeprest_amd.engine.config import DataConfig, EngineConfig, TrainConfig
from deeprest_amd.engine.experiment import run_experiment
from deeprest_amd.engine.trainer import Trainer
from deeprest_amd.models.net import DeepRestNetConfig
from deeprest_amd.serve.anomaly import AnomalyScorer
from deeprest_amd.serve.predictor import Predictor


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpu", action="store_true")
    ap.add_argument("--epochs", type=int, default=3)
    args = ap.parse_args()
    device = torch.device("cuda" if args.gpu else "cpu")

    print("== 1. synthetic application (replaces the social-network testbed)")
    app = SyntheticApp(SyntheticAppConfig(
        n_apis=8, n_components=12, windows_per_day=240, n_days=3, seed=42))
    raw = app.generate_raw()
    print(f"   {len(raw)} windows, {len(app.apis)} API endpoints, "
          f"{len(app.all_components)} components")

    print("== 2. featurize (call-path feature space)")
    data = Featurizer().fit_transform(raw)
    print(f"   {data.num_paths} call paths, {len(data.metric_names)} metrics")

    print("== 3. train with the RESRC/COMP/DEEPR comparison harness")
    cfg = EngineConfig(
        data=DataConfig(step_size=60, split=0.4),
        train=TrainConfig(epochs=args.epochs, batch_size=32, baseline_epochs=20,
                          log_every=1, checkpoint_path="/tmp/deeprest_amd_ckpt.pt",
                          dtype="bf16" if args.gpu else "fp32"),
        model=DeepRestNetConfig(
            d_model=128, n_heads=4, n_layers=1, d_ff=256,
            hidden=128 if args.gpu else 32,  # GPU kernel is H=128-native
            comp_dim=32, dropout=0.1),
    )
    trainer = Trainer(data, cfg, device=device)
    result = trainer.train()
    print(f"   {result.samples_per_sec:.0f} windows/s")

    print("== 4. what-if estimation for an unseen traffic mix")
    syn = TraceSynthesizer(feature_space=data.feature_space).fit(raw)
    pred = Predictor.from_checkpoint("/tmp/deeprest_amd_ckpt.pt", device=device,
                                     use_graph=args.gpu)
    plan = [{app.apis[0]: 40, app.apis[1]: 10}] * 120  # hypothetical mix
    what_if = pred.predict_what_if(syn, plan, step_size=60,
                                   rng=np.random.default_rng(0))
    name = data.metric_names[0]
    q = what_if[name][0]   # (T, 3) quantiles of the first window
    print(f"   {name}: median ~{q[:, 1].mean():.1f} "
          f"(band {q[:, 0].mean():.1f}..{q[:, 2].mean():.1f})")

    print("== 5. sanity check: inject a cryptojacking burst and detect it")
    comp = app.components[0]
    app.inject_anomaly(data, comp, "cpu", start=500, length=40, magnitude=4.0)
    series = data.resources[f"{comp}_cpu"]
    from deeprest_amd.data.windows import sliding_window
    w = sliding_window(data.traffic.astype(np.float64), 60)
    preds = pred.predict(w[::60][:10])
    # score the injected region against the predicted band
    m_idx = data.metric_names.index(f"{comp}_cpu")
    scorer = AnomalyScorer()
    win_idx = 500 // 60  # which eval window holds the burst
    if win_idx < len(preds[f"{comp}_cpu"]):
        band = preds[f"{comp}_cpu"][win_idx]
        seg = series[win_idx * 60:(win_idx + 1) * 60][:band.shape[0]]
        rep = scorer.score(seg, band[:, 0], band[:, 1], band[:, 2],
                           metric=f"{comp}_cpu")
        print(f"   anomalous={rep.is_anomalous} windows={rep.windows}")

    print("== 6. full experiment -> results.pkl (web-demo schema)")
    cfg.train.epochs = 1
    store = run_experiment(data, "quickstart-waves_seen-1x", config=cfg,
                           device=device)
    store.save("/tmp/deeprest_amd_results.pkl")
    print("   saved /tmp/deeprest_amd_results.pkl "
          f"({len(store.results['quickstart-waves_seen-1x'])} components)")


if __name__ == "__main__":
    main()
