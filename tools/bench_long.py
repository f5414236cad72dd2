#!/usr/bin/env python3
"""Long-horizon streaming inference benchmark (BASELINE config 5).

Streams a multi-day 1s-granularity horizon through the chunked encoder +
state-carrying GRU decode (models/net.py forward_long), optionally on the
fp8 MFMA path, and reports horizon steps/sec and peak memory.

  python tools/bench_long.py [--horizon 604800] [--fp8] [--endpoints 256]
"""
import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import torch

from deeprest_amd.data.synthetic import SyntheticApp, SyntheticAppConfig
from deeprest_amd.models.net import DeepRestNet, DeepRestNetConfig, build_model_spec


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--horizon", type=int, default=604_800, help="total steps (7d@1s)")
    ap.add_argument("--chunk", type=int, default=4096)
    ap.add_argument("--endpoints", type=int, default=256)
    ap.add_argument("--components", type=int, default=63)
    ap.add_argument("--fp8", action="store_true")
    ap.add_argument("--iters", type=int, default=1)
    args = ap.parse_args()

    dev = torch.device("cuda" if torch.cuda.is_available() else "cpu")
    app = SyntheticApp(SyntheticAppConfig(
        n_apis=args.endpoints, n_components=args.components,
        windows_per_day=240, n_days=1, seed=7))
    data = app.generate_featurized()
    spec = build_model_spec(data)
    torch.manual_seed(0)
    model = DeepRestNet(spec, DeepRestNetConfig(
        dropout=0.0, fp8_inference=args.fp8)).to(dev).eval()

    P = spec.num_paths
    # synthesize the horizon chunk-by-chunk on device to bound host memory:
    # use one random traffic tensor re-used across chunks (shape evidence only)
    x_chunk = torch.rand(1, args.chunk, P, device=dev)
    n_chunks = (args.horizon + args.chunk - 1) // args.chunk

    # emulate forward_long's streaming loop over a horizon of repeated chunks
    with torch.no_grad():
        # warmup
        model.forward_long(x_chunk, chunk_size=args.chunk)
        if dev.type == "cuda":
            torch.cuda.synchronize()
            torch.cuda.reset_peak_memory_stats()
        t0 = time.perf_counter()
        comp = model.graph()
        dec = model.decoder
        H = dec.hidden
        h_f = torch.tanh(dec.h0_proj(comp)).unsqueeze(0).expand(1, comp.shape[0], H).contiguous()
        gamma = dec.cond_gamma(comp)
        beta = dec.cond_beta(comp)
        from deeprest_amd.ops import fused_gru_sequence

        done = 0
        for ci in range(n_chunks):
            enc = model.in_proj(x_chunk)
            enc = enc + model._pos_encoding(args.chunk, dev, enc.dtype)
            enc = model.in_norm(enc)
            for layer in model.layers:
                enc = layer(enc)
            xgc = dec.x_proj(enc)
            out = fused_gru_sequence(xgc, dec.w_hh, dec.b_hh, h_f, gamma, beta,
                                     reverse=False, fp8=args.fp8)
            h_f = out[:, -1].contiguous()
            done += args.chunk
        if dev.type == "cuda":
            torch.cuda.synchronize()
        dt = time.perf_counter() - t0

    peak_gb = (torch.cuda.max_memory_allocated() / 2**30) if dev.type == "cuda" else 0.0
    print(json.dumps({
        "metric": "long-horizon streaming decode",
        "horizon_steps": done,
        "seconds": round(dt, 3),
        "steps_per_sec": round(done / dt, 1),
        "chunk": args.chunk,
        "fp8": args.fp8,
        "endpoints": args.endpoints,
        "num_paths": P,
        "peak_mem_gb": round(peak_gb, 2),
    }))


if __name__ == "__main__":
    main()
