#!/usr/bin/env python3
"""Narrow the graph-replay NaN further.

Variants:
  graph      — full captured step (repro: NaN at epoch 7)
  graphfb    — capture forward+backward only, Adam steps eagerly
  graphdrop  — full captured step, drop_last (no eager tail batches)
  graphsync  — full captured step + per-step torch.cuda.synchronize()

Per-step finiteness check pinpoints the exact failing step.
"""
import argparse
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import torch

from deeprest_amd.data.synthetic import SyntheticApp, SyntheticAppConfig
from deeprest_amd.engine.dataset import EstimationDataset
from deeprest_amd.engine.graphstep import GraphedTrainStep
from deeprest_amd.models.net import DeepRestNet, DeepRestNetConfig, build_model_spec
from deeprest_amd.ops.adam import FusedAdam


class FwdBwdGraph:
    """Capture only zero_grad+forward+loss+backward; optimizer stays eager."""

    def __init__(self, model, opt, loss_fn, xb, yb):
        self.model, self.loss_fn = model, loss_fn
        self.static_x = xb.clone()
        self.static_y = yb.clone()
        self.opt = opt
        side = torch.cuda.Stream()
        side.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(side):
            for _ in range(3):
                self._one()
                opt.step()
        torch.cuda.current_stream().wait_stream(side)
        self.g = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self.g, stream=side):
            self.static_loss = self._one()

    def _one(self):
        with torch.autocast("cuda", torch.bfloat16):
            out = self.model(self.static_x)
            loss = self.loss_fn(out.float(), self.static_y)
        self.opt.zero_grad(set_to_none=False)
        loss.backward()
        return loss

    def run(self, xb, yb):
        self.static_x.copy_(xb)
        self.static_y.copy_(yb)
        self.g.replay()
        self.opt.step()
        return self.static_loss


def run_variant(variant, data, device, epochs, seed):
    torch.manual_seed(seed)
    ds = EstimationDataset(data, step_size=60, split_fraction=0.40)
    spec = build_model_spec(data)
    model = DeepRestNet(spec, DeepRestNetConfig(dropout=0.0)).to(device)
    opt = FusedAdam(model.parameters(), lr=1e-3,
                    capturable=variant != "syncplainadam")
    loss_fn = lambda o, t: model.loss(o, t)
    X = ds.X_train.to(device)
    y = ds.y_train.to(device)
    n = X.shape[0]
    bs = 32
    gen = torch.Generator().manual_seed(seed)
    g = None
    step_i = 0
    sync = variant.startswith("sync") or variant == "graphsync"
    for epoch in range(epochs):
        perm = torch.randperm(n, generator=gen)
        drop = variant in ("graphdrop", "syncdrop", "syncfb", "syncplainadam")
        stop = n - (n % bs) if drop else n
        losses, steps = [], []
        for s in range(0, stop, bs):
            idx = perm[s : s + bs].to(device)
            xb, yb = X[idx], y[idx]
            if g is None and xb.shape[0] == bs:
                if variant in ("graphfb", "syncfb", "syncplainadam"):
                    g = FwdBwdGraph(model, opt, loss_fn, xb, yb)
                else:
                    g = GraphedTrainStep(
                        model, opt,
                        lambda o, t: model.loss(o.float(), t), xb, yb,
                        autocast_dtype=torch.bfloat16)
            if g is not None and xb.shape[0] == bs:
                loss = g.run(xb, yb)
            else:  # eager tail
                cache = variant != "synccachefree"
                with torch.autocast("cuda", torch.bfloat16,
                                    cache_enabled=cache):
                    out = model(xb)
                    loss = model.loss(out.float(), yb)
                opt.zero_grad(set_to_none=False)
                loss.backward()
                opt.step()
            if sync:
                torch.cuda.synchronize()
            losses.append(loss.detach().clone())
            steps.append((step_i, xb.shape[0]))
            step_i += 1
        # epoch-end check keeps steps async (a per-step host sync could
        # mask a replay race); pinpoints the first bad step after the fact
        lv = [float(x) for x in torch.stack(losses)]
        bad = [i for i, x in enumerate(lv) if x != x]
        if bad:
            print(f"  !! NaN at epoch {epoch}, first bad batch index "
                  f"{bad[0]}/{len(lv)} (global step {steps[bad[0]][0]}, "
                  f"batch size {steps[bad[0]][1]})", flush=True)
            return epoch
        if epoch % 5 == 0:
            print(f"  epoch {epoch}: {lv[-1]:.4f}", flush=True)
    return None


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--epochs", type=int, default=12)
    ap.add_argument("--variants", default="graph,graphfb,graphdrop,graphsync")
    args = ap.parse_args()
    device = torch.device("cuda")
    app = SyntheticApp(SyntheticAppConfig(
        n_apis=13, n_components=12, windows_per_day=240, n_days=8,
        resource_noise=0.03, seed=77))
    data = app.generate_featurized()
    for variant in args.variants.split(","):
        print(f"variant={variant}", flush=True)
        bad = run_variant(variant, data, device, args.epochs, 0)
        print(f"variant={variant} -> "
              f"{'NaN@' + str(bad) if bad is not None else 'ok'}", flush=True)


if __name__ == "__main__":
    main()
