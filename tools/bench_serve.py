#!/usr/bin/env python3
"""Online-prediction latency benchmark (BASELINE config 3): batched 1k-window
inference through the hipGraph-captured predictor; reports p50/p95 latency.

Run on a GPU box:  python tools/bench_serve.py [--windows 1000] [--iters 30]
"""
import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import numpy as np
import torch

from deeprest_amd.data.synthetic import SyntheticApp, SyntheticAppConfig
from deeprest_amd.data.windows import MinMaxScaler, sliding_window
from deeprest_amd.models.net import DeepRestNet, DeepRestNetConfig, build_model_spec
from deeprest_amd.serve.predictor import Predictor


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--windows", type=int, default=1000)
    p.add_argument("--iters", type=int, default=30)
    p.add_argument("--warmup", type=int, default=5)
    p.add_argument("--endpoints", type=int, default=256)
    p.add_argument("--components", type=int, default=64)
    p.add_argument("--seq-len", type=int, default=60)
    p.add_argument("--graph-batch", type=int, default=256)
    p.add_argument("--no-graph", action="store_true")
    args = p.parse_args()

    dev = torch.device("cuda" if torch.cuda.is_available() else "cpu")
    app = SyntheticApp(SyntheticAppConfig(
        n_apis=args.endpoints, n_components=args.components,
        windows_per_day=max(240, args.seq_len * 4 + 8), n_days=1, seed=7))
    data = app.generate_featurized()
    spec = build_model_spec(data)
    torch.manual_seed(0)
    model = DeepRestNet(spec, DeepRestNetConfig(dropout=0.0)).to(dev).eval()
    x_scaler = MinMaxScaler().fit(data.traffic.astype(np.float64), 200)
    y_scalers = [MinMaxScaler() for _ in data.metric_names]
    pred = Predictor(model, x_scaler, y_scalers, data.metric_names, device=dev,
                     graph_batch=args.graph_batch, use_graph=not args.no_graph)

    w = sliding_window(data.traffic.astype(np.float64), args.seq_len)
    reps = int(np.ceil(args.windows / len(w)))
    w = np.concatenate([w] * reps)[: args.windows]
    # resident on device: the benchmark measures the prediction path, not PCIe
    xn = torch.from_numpy(x_scaler.transform(w)).float().to(dev)

    def time_requests(request_windows):
        lat = []
        for i in range(args.warmup + args.iters):
            if dev.type == "cuda":
                torch.cuda.synchronize()
            t0 = time.perf_counter()
            pred.predict_normalized(request_windows)
            if dev.type == "cuda":
                torch.cuda.synchronize()
            dt = time.perf_counter() - t0
            if i >= args.warmup:
                lat.append(dt * 1000.0)
        return np.asarray(lat)

    # bulk: one 1k-window batch
    lat = time_requests(xn)
    # small latency-sensitive request (graph batch matches it)
    small = xn[: min(16, len(xn))]
    if not args.no_graph:
        pred_small = Predictor(model, x_scaler, y_scalers, data.metric_names,
                               device=dev, graph_batch=len(small), use_graph=True)
    else:
        pred_small = pred
    lat_s = []
    for i in range(args.warmup + args.iters):
        if dev.type == "cuda":
            torch.cuda.synchronize()
        t0 = time.perf_counter()
        pred_small.predict_normalized(small)
        if dev.type == "cuda":
            torch.cuda.synchronize()
        if i >= args.warmup:
            lat_s.append((time.perf_counter() - t0) * 1000.0)
    lat_s = np.asarray(lat_s)
    print(json.dumps({
        "metric": "online inference latency",
        "bulk_p50_ms": round(float(np.percentile(lat, 50)), 3),
        "bulk_p95_ms": round(float(np.percentile(lat, 95)), 3),
        "bulk_windows": args.windows,
        "bulk_windows_per_sec": round(args.windows / (np.percentile(lat, 50) / 1000.0), 1),
        "small_request_windows": int(len(small)),
        "small_p50_ms": round(float(np.percentile(lat_s, 50)), 3),
        "small_p95_ms": round(float(np.percentile(lat_s, 95)), 3),
        "hipgraph": pred.use_graph,
        "graph_batch": args.graph_batch,
        "endpoints": args.endpoints,
        "seq_len": args.seq_len,
    }))


if __name__ == "__main__":
    main()
