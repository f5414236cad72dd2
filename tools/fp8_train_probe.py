#!/usr/bin/env python3
"""fp8-activation training experiment (BASELINE config 5 / VERDICT item 4).

Question: can training run with fp8 (e4m3) activations feeding the GRU
decoders and the attention QKV — the two MFMA consumers — without losing
accuracy?  This probe fake-quantizes those activations (cast to
float8_e4m3fnuz and back, forward-only; gradients flow through the
straight-through estimator) and compares the 50-epoch reference-config
error table against the bf16 baseline.  The outcome decides whether a
true fp8 training kernel path is worth building.

  python tools/fp8_train_probe.py [--epochs 50]
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

FP8 = getattr(torch, "float8_e4m3fnuz", None) or torch.float8_e4m3fn


class _FakeQuant(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x):
        # per-tensor dynamic scale to the e4m3 range, like an fp8 GEMM's
        # scaling factor would apply.  NOTE: the ROCm variant is
        # e4m3fnuz whose max finite value is 240 (not OCP e4m3's 448) —
        # scaling to 448 overflowed to NaN
        fmax = float(torch.finfo(FP8).max) * 0.98
        amax = x.detach().abs().amax().clamp(min=1e-8)
        scale = fmax / amax
        return (x * scale).to(FP8).to(x.dtype) / scale

    @staticmethod
    def backward(ctx, g):
        return g  # straight-through


def add_fp8_hooks(model):
    """Quantize the activations entering the MFMA consumers."""
    hooks = []

    def q(_mod, _inp, out):
        return _FakeQuant.apply(out)

    dec = model.decoder
    for mod in [dec.x_proj, getattr(dec, "x_proj_r", None)]:
        if mod is not None:
            hooks.append(mod.register_forward_hook(q))
    for layer in model.layers:
        hooks.append(layer.qkv.register_forward_hook(q))
    return hooks


def run(epochs, fp8, device, seed=77):
    app = SyntheticApp(SyntheticAppConfig(
        n_apis=13, n_components=12, windows_per_day=240, n_days=8,
        resource_noise=0.03, seed=seed))
    data = app.generate_featurized()
    cfg = EngineConfig(
        data=DataConfig(step_size=60, split=0.40),
        train=TrainConfig(epochs=epochs, batch_size=32, lr=1e-3,
                          eval_cycles=9, baseline_epochs=100, log_every=0,
                          eval_every=5),
        model=DeepRestNetConfig(dropout=0.1),
    )
    torch.manual_seed(0)
    trainer = Trainer(data, cfg, device=device)
    if fp8:
        add_fp8_hooks(trainer.model)
    result = trainer.train()
    med = {"resrc": [], "comp": [], "deepr": []}
    for per_est in result.error_tables.values():
        for k in med:
            med[k].append(per_est[k]["median"])
    return {k: round(float(np.mean(v)), 4) for k, v in med.items()}


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--epochs", type=int, default=50)
    args = ap.parse_args()
    device = torch.device("cuda" if torch.cuda.is_available() else "cpu")
    bf16 = run(args.epochs, fp8=False, device=device)
    fp8 = run(args.epochs, fp8=True, device=device)
    print(f"fp8 dtype: {FP8}")
    print(f"bf16 activations : {bf16}")
    print(f"fp8  activations : {fp8}")
    delta = {k: round(fp8[k] - bf16[k], 4) for k in bf16}
    rel = round((fp8["deepr"] - bf16["deepr"]) / max(bf16["deepr"], 1e-9) * 100, 1)
    print(f"delta            : {delta}  (DEEPR {rel:+.1f}%)")


if __name__ == "__main__":
    main()
