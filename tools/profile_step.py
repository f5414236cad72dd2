#!/usr/bin/env python3
"""Op-level training-step profile with input shapes (torch.profiler).

rocprofv3 gives per-kernel time but anonymous hipBLASLt kernel names make it
hard to attribute GEMM time to model ops; this runs a few bench-config
training steps under torch.profiler with record_shapes and prints the top
ops by device time — the map from `Cijk_*` kernels back to "whose GEMM is
that".

  python tools/profile_step.py [--batch 512] [--steps 3] [--endpoints 256]
"""
import argparse
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import torch

from deeprest_amd.data.synthetic import SyntheticApp, SyntheticAppConfig
from deeprest_amd.engine.dataset import EstimationDataset
from deeprest_amd.models.net import DeepRestNet, DeepRestNetConfig, build_model_spec
from deeprest_amd.ops.adam import FusedAdam


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--batch", type=int, default=512)
    ap.add_argument("--endpoints", type=int, default=256)
    ap.add_argument("--components", type=int, default=63)
    ap.add_argument("--seq-len", type=int, default=60)
    ap.add_argument("--steps", type=int, default=3)
    ap.add_argument("--row-limit", type=int, default=40)
    args = ap.parse_args()

    dev = torch.device("cuda" if torch.cuda.is_available() else "cpu")
    T = args.seq_len
    need = int(args.batch / 0.8) + T + 64
    app = SyntheticApp(SyntheticAppConfig(
        n_apis=args.endpoints, n_components=args.components,
        windows_per_day=max(4 * T, need), n_days=1, seed=1234))
    data = app.generate_featurized()
    ds = EstimationDataset(data, step_size=T, split_fraction=0.8)
    spec = build_model_spec(data)
    torch.manual_seed(0)
    model = DeepRestNet(spec, DeepRestNetConfig(dropout=0.0)).to(dev)
    opt = FusedAdam(model.parameters(), lr=1e-3)
    X = ds.X_train[: args.batch].to(dev)
    y = ds.y_train[: args.batch].to(dev)
    if dev.type == "cuda":
        X = X.to(torch.bfloat16)

    def step():
        with torch.autocast(device_type="cuda", dtype=torch.bfloat16,
                            enabled=dev.type == "cuda"):
            out = model(X)
            loss = model.loss(out.float(), y)
        opt.zero_grad(set_to_none=True)
        loss.backward()
        opt.step()

    for _ in range(2):
        step()
    if dev.type == "cuda":
        torch.cuda.synchronize()

    acts = [torch.profiler.ProfilerActivity.CPU]
    if dev.type == "cuda":
        acts.append(torch.profiler.ProfilerActivity.CUDA)
    with torch.profiler.profile(activities=acts, record_shapes=True) as prof:
        for _ in range(args.steps):
            step()
        if dev.type == "cuda":
            torch.cuda.synchronize()

    sort_key = "self_cuda_time_total" if dev.type == "cuda" else "self_cpu_time_total"
    print(prof.key_averages(group_by_input_shape=True).table(
        sort_by=sort_key, row_limit=args.row_limit, max_src_column_width=60))


if __name__ == "__main__":
    main()
