#!/usr/bin/env python3
"""Online-prediction latency benchmark (BASELINE config 3).

Measures p50/p95 request latency for the graph-captured predictor vs the
eager path at request sizes 16 / 256 / 1024 windows, plus the staged-
ingestion variant (windows written directly into the graph's input buffer).

Run on a GPU box:  python tools/bench_serve.py [--iters 30]
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
    p.add_argument("--iters", type=int, default=30)
    p.add_argument("--warmup", type=int, default=5)
    p.add_argument("--endpoints", type=int, default=256)
    p.add_argument("--components", type=int, default=64)
    p.add_argument("--seq-len", type=int, default=60)
    p.add_argument("--sizes", default="16,256,1024")
    args = p.parse_args()
    sizes = [int(s) for s in args.sizes.split(",")]

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
    graphed = Predictor(model, x_scaler, y_scalers, data.metric_names,
                        device=dev, graph_batches=tuple(sizes), use_graph=True)
    eager = Predictor(model, x_scaler, y_scalers, data.metric_names,
                      device=dev, use_graph=False)

    w = sliding_window(data.traffic.astype(np.float64), args.seq_len)
    reps = int(np.ceil(max(sizes) / len(w)))
    w = np.concatenate([w] * reps)[: max(sizes)]
    xn_cpu = torch.from_numpy(x_scaler.transform(w)).float()
    xn_dev = xn_cpu.to(dev)
    T, P = xn_dev.shape[1], xn_dev.shape[2]

    def timed(fn):
        lat = []
        for i in range(args.warmup + args.iters):
            if dev.type == "cuda":
                torch.cuda.synchronize()
            t0 = time.perf_counter()
            fn()
            if dev.type == "cuda":
                torch.cuda.synchronize()
            if i >= args.warmup:
                lat.append((time.perf_counter() - t0) * 1000.0)
        a = np.asarray(lat)
        return {"p50_ms": round(float(np.percentile(a, 50)), 3),
                "p95_ms": round(float(np.percentile(a, 95)), 3)}

    results = {"device": str(dev), "endpoints": args.endpoints,
               "num_paths": spec.num_paths, "seq_len": args.seq_len,
               "iters": args.iters, "sizes": {}}
    for n in sizes:
        req_dev = xn_dev[:n]
        entry = {}
        entry["eager"] = timed(lambda: eager.predict_normalized(req_dev))
        entry["graph"] = timed(lambda: graphed.predict_normalized(req_dev))

        def staged():
            buf = graphed.staging_buffer(n, T, P)
            buf.copy_(req_dev)
            graphed.predict_staged(n, T)
        entry["graph_staged"] = timed(staged)
        # streamed-ingestion model: windows already landed in the staging
        # buffer during ingestion, the request itself is one replay
        graphed.staging_buffer(n, T, P).copy_(req_dev)
        entry["graph_prestaged"] = timed(lambda: graphed.predict_staged(n, T))
        results["sizes"][n] = entry
        print(f"n={n}: {json.dumps(entry)}", flush=True)
    print(json.dumps(results))


if __name__ == "__main__":
    main()
