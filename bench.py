#!/usr/bin/env python3
"""Flagship training-step benchmark (driver contract).

Measures the BASELINE.json headline metric: training samples(windows)/sec of
the 256-endpoint / 3-resource estimation model, weak-scaled over 1..8 MI355X
GPUs (per-GPU batch fixed).  Synthetic traffic of the named shape, random
init weights (no network access for datasets), full training step timed:
forward + pinball loss + backward + gradient all-reduce + fused Adam step.
The step itself is engine/step.py's TrainStep — the exact code path
Trainer runs, not a bench-only reimplementation.

After the timed region (single-process GPU runs) a 50-epoch accuracy probe
at the reference training config (50 epochs / batch 32 / split .40 /
window 60 — reference: resource-estimation/estimate.py:13-18) reports the
prediction-MAE half of the metric: mean median absolute error for the
RESRC / COMP / DEEPR estimators (reference: estimate.py:112-122).

Usage (driver):
  python bench.py --gpus N --steps K --warmup W
  torchrun --nnodes=1 --nproc-per-node N bench.py --gpus N --steps K --warmup W

Rank 0 prints exactly one JSON line with the whole-job aggregate.
"""

from __future__ import annotations

import argparse
import hashlib
import json
import os
import sys
import tempfile
import time

# Tuned hipBLASLt/rocBLAS algorithm tables for this image (ROCm TunableOp;
# measured +4.5% windows/s at the flagship config — the default heuristics
# leave the skinny-N in_proj GEMMs on slow algorithms).  Must be exported
# before torch initializes; TUNING=0 means replay-only (shapes missing
# from the table silently use the default algorithm).
_TUNED = os.path.join(os.path.dirname(os.path.abspath(__file__)),
                      "configs", "tunableop.csv")
if os.path.exists(_TUNED.replace(".csv", "0.csv")):
    os.environ.setdefault("PYTORCH_TUNABLEOP_ENABLED", "1")
    os.environ.setdefault("PYTORCH_TUNABLEOP_TUNING", "0")
    os.environ.setdefault("PYTORCH_TUNABLEOP_FILENAME", _TUNED)

import torch

from deeprest_amd.data.synthetic import SyntheticApp, SyntheticAppConfig
from deeprest_amd.engine.dataset import EstimationDataset
from deeprest_amd.engine.step import TrainStep
from deeprest_amd.models.net import DeepRestNet, DeepRestNetConfig, build_model_spec
from deeprest_amd.ops.adam import FusedAdam
from deeprest_amd.parallel.dist import init_distributed


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=30)
    p.add_argument("--warmup", type=int, default=10)
    # 288 GB of HBM3E per GPU sizes the window batch (north star): 2048
    # keeps rows a multiple of the 64x256 CU tiling and measured +10%
    # windows/s over 1024 (48.6k vs 44.0k, tuned GEMMs); rank-0-builds +
    # /dev/shm-mmap dataset sharing keeps 8-rank host RAM at one copy
    p.add_argument("--batch", type=int, default=2048, help="per-GPU window batch")
    p.add_argument("--endpoints", type=int, default=256, help="API endpoints")
    # 63 + the frontend component = 64 -> batch*64 rows tile exactly onto the
    # 256 CUs for the fused GRU kernels (a 260th workgroup at 1 block/CU
    # doubles their wall time)
    p.add_argument("--components", type=int, default=63)
    p.add_argument("--seq-len", type=int, default=60)
    p.add_argument("--dtype", type=str, default="bf16", choices=["bf16", "fp32"])
    p.add_argument("--device", type=str, default=None)
    # MAE half of the headline metric: auto = run on single-process GPU
    # benches (the driver's N=1 run), skip in scale runs and on CPU
    p.add_argument("--accuracy", choices=["auto", "on", "off"], default="auto")
    p.add_argument("--accuracy-epochs", type=int, default=50)
    return p.parse_args()


def build_train_tensors(args):
    """Synthetic 256-endpoint app -> train split tensors + model spec."""
    T = args.seq_len
    # enough raw windows that the train split holds >= one full batch
    need = int(args.batch / 0.8) + T + 64
    app = SyntheticApp(SyntheticAppConfig(
        n_apis=args.endpoints,
        n_components=args.components,
        windows_per_day=max(4 * T, need),
        n_days=1,
        shapes_per_api=3,
        seed=1234,
    ))
    data = app.generate_featurized()
    ds = EstimationDataset(data, step_size=T, split_fraction=0.8)
    spec = build_model_spec(data)
    # clone: X_train is a view — saving/keeping it would pin the whole
    # (train+test) storage
    X, y = ds.X_train.clone(), ds.y_train.clone()
    pad = (-spec.num_paths) % 64
    if pad and os.environ.get("DEEPREST_PAD64", "1") == "1":
        # 128-byte row alignment for the in_proj GEMMs: an odd lda
        # (12615 call paths) makes every A-row read cross cache lines
        # unaligned; zero columns are inert under the global min-max
        # (traffic min is 0) and the padded K is what the shipped
        # TunableOp tables are keyed on
        X = torch.nn.functional.pad(X, (0, pad))
        spec.num_paths += pad
    del ds, data, app  # free the multi-GB host transients
    return X, y, spec


def shared_train_tensors(args, dist_ctx):
    """Rank 0 builds the dataset ONCE and the other ranks mmap it from shm:
    8 ranks each constructing an identical multi-GB synthetic dataset was
    ~8x the host RAM and a serial startup cost inside the driver's clock."""
    if dist_ctx is None:
        return build_train_tensors(args)
    key = f"{args.batch}-{args.endpoints}-{args.components}-{args.seq_len}"
    # estimated file size: X f32 (need, T, ~P) + slack; containers often
    # mount a small /dev/shm — fall back to TMPDIR rather than ENOSPC the
    # driver's 8-rank scale run
    need_windows = int(args.batch / 0.8) + args.seq_len + 64
    est_bytes = int(need_windows * args.seq_len
                    * (args.endpoints * 55 + 64) * 4 * 1.3)
    base = tempfile.gettempdir()
    if os.path.isdir("/dev/shm"):
        st = os.statvfs("/dev/shm")
        if st.f_bavail * st.f_frsize > est_bytes:
            base = "/dev/shm"
    path = os.path.join(
        base, f"deeprest_bench_{hashlib.md5(key.encode()).hexdigest()[:12]}.pt")
    if dist_ctx.rank == 0:
        X, y, spec = build_train_tensors(args)
        tmp = f"{path}.tmp{os.getpid()}"
        torch.save({"X": X, "y": y, "spec": spec}, tmp)
        os.replace(tmp, path)
        dist_ctx.barrier()          # file is in place -> others may read
    else:
        dist_ctx.barrier()
        # trusted file: rank 0 of this same launch just wrote it
        blob = torch.load(path, map_location="cpu", mmap=True,
                          weights_only=False)
        X, y, spec = blob["X"], blob["y"], blob["spec"]
    dist_ctx.barrier()              # everyone mapped -> safe to unlink
    if dist_ctx.rank == 0:
        try:
            os.unlink(path)
        except OSError:
            pass
    return X, y, spec


def accuracy_probe(epochs: int, device: torch.device, n_apis: int = 13,
                   n_components: int = 12, windows_per_day: int = 240,
                   n_days: int = 8, step_size: int = 60,
                   baseline_epochs: int = 100,
                   lr_schedule: str = "none") -> dict:
    """Train at the reference config on a reference-scale app and report the
    three-estimator mean median absolute error (the prediction-MAE half of
    BASELINE.json's metric; anchors: resource-estimation/README.md:88-98)."""
    import numpy as np

    from deeprest_amd.engine.config import DataConfig, EngineConfig, TrainConfig
    from deeprest_amd.engine.trainer import Trainer

    t0 = time.perf_counter()
    app = SyntheticApp(SyntheticAppConfig(
        n_apis=n_apis, n_components=n_components,
        windows_per_day=windows_per_day, n_days=n_days,
        resource_noise=0.03, seed=77))
    data = app.generate_featurized()

    def train_once(n_epochs):
        cfg = EngineConfig(
            data=DataConfig(step_size=step_size, split=0.40),
            train=TrainConfig(epochs=n_epochs, batch_size=32, lr=1e-3,
                              eval_cycles=9, baseline_epochs=baseline_epochs,
                              log_every=0, eval_every=5, graph_step=True,
                              lr_schedule=lr_schedule,
                              # flagship estimator: net + trace-ridge
                              # residual base (A/B: DEEPR 3.9 vs 7.8 at 50
                              # epochs, beats COMP 39/39) with conformal
                              # band calibration (coverage 0.90 nominal)
                              residual_base="trace-ridge", conformal=0.9),
            model=DeepRestNetConfig(dropout=0.1),
        )
        torch.manual_seed(0)  # probe stability run-to-run
        trainer = Trainer(data, cfg, device=device)
        return trainer, trainer.train()

    def aggregate(result):
        wins_resrc = wins_comp = total = 0
        med = {"resrc": [], "comp": [], "deepr": []}
        for per_est in result.error_tables.values():
            total += 1
            for k in med:
                med[k].append(per_est[k]["median"])
            wins_resrc += per_est["deepr"]["median"] <= per_est["resrc"]["median"]
            wins_comp += per_est["deepr"]["median"] <= per_est["comp"]["median"]
        return ({k: round(float(np.mean(v)), 4) for k, v in med.items()},
                int(wins_resrc), int(wins_comp), total)

    trainer, result = train_once(epochs)
    t_ref = time.perf_counter() - t0
    med, wins_resrc, wins_comp, total = aggregate(result)
    cov = [c["coverage"] for c in result.coverage.values()]
    out = {
        "mean_median_abs_err": med,
        "deepr_beats_resrc": wins_resrc,
        "deepr_beats_comp": wins_comp,
        "metrics": total,
        "epochs": epochs,
        # empirical coverage of the conformalized (.05,.95) band
        # (nominal 0.90)
        "band_coverage": round(float(np.mean(cov)), 4) if cov else None,
        "config": "reference 50ep/b32/split.40/window60 config, "
                  "13-endpoint 12-component 8-day synthetic app",
    }
    # extended run: the model keeps converging past the reference's 50
    # epochs (measured: DEEPR ~4.8 vs COMP 9.97 at 100 epochs); report it
    # next to the reference-config row without replacing it
    _, result2 = train_once(2 * epochs)
    med2, _, wins_comp2, _ = aggregate(result2)
    out["extended"] = {"epochs": 2 * epochs,
                       "deepr": med2["deepr"],
                       "deepr_beats_comp": wins_comp2}
    out["probe_seconds"] = round(time.perf_counter() - t0, 1)
    out["reference_config_seconds"] = round(t_ref, 1)
    return out


def main():
    args = parse_args()
    dist_ctx = init_distributed()
    rank = dist_ctx.rank if dist_ctx else 0
    world = dist_ctx.world_size if dist_ctx else 1

    on_gpu = torch.cuda.is_available()
    if args.device:
        device = torch.device(args.device)
    elif on_gpu:
        # dist_ctx.device already maps local_rank modulo the visible GPU
        # count (a 2-rank RCCL smoke on a 1-GPU box shares device 0)
        device = dist_ctx.device if dist_ctx else torch.device("cuda", 0)
        torch.cuda.set_device(device)
    else:
        device = torch.device("cpu")

    X, y, spec = shared_train_tensors(args, dist_ctx)
    X = X.to(device)
    y = y.to(device)
    T = args.seq_len

    torch.manual_seed(1234)
    model = DeepRestNet(spec, DeepRestNetConfig(dropout=0.0)).to(device)
    if dist_ctx is not None:
        # belt and braces on top of the seeded construction: replicas MUST
        # start identical or averaged gradients never reconcile them
        dist_ctx.broadcast_parameters(model)
    opt = FusedAdam(model.parameters(), lr=1e-3, capturable=on_gpu)

    n = X.shape[0]
    B = args.batch
    autocast_dtype = torch.bfloat16 if args.dtype == "bf16" else torch.float32
    use_autocast = on_gpu and args.dtype == "bf16"
    if use_autocast:
        # autocast would cast the (B, T, P) traffic input to bf16 EVERY step
        # before the in_proj GEMM (~1.5 GB at this config); storing it bf16
        # once is bit-identical to what the per-step cast feeds the GEMM
        X = X.to(torch.bfloat16)

    if n < B:
        raise RuntimeError(f"only {n} train windows for per-GPU batch {B}")

    step = TrainStep(model, opt, dist_ctx=dist_ctx,
                     autocast_dtype=autocast_dtype if use_autocast else None)
    # hipGraph training (DEEPREST_GRAPH_STEP=1, single-process): capture K
    # graphs over K persistent SLICES of the resident training set, sharing
    # one capture pool, and cycle replays — zero input copies (round 1's
    # single-graph variant lost because every replay paid a ~1.5 GB copy
    # into its static buffer) and no per-launch gaps.
    graph_cycle = None
    if (on_gpu and world == 1
            and os.environ.get("DEEPREST_GRAPH_STEP", "0") == "1"):
        from deeprest_amd.engine.graphstep import GraphedTrainStep

        K = int(os.environ.get("DEEPREST_GRAPH_SLICES", "4"))
        offs = sorted({((k * B + 37 * k) % max(n - B, 1)) for k in range(K)})
        cycle = []
        pool = None
        try:
            for o in offs:
                g = GraphedTrainStep(
                    model, opt, lambda out, t: model.loss(out, t),
                    X[o : o + B], y[o : o + B],
                    autocast_dtype=autocast_dtype if use_autocast else None,
                    warmup=2, static_inputs=True, pool=pool)
                pool = pool or g.graph.pool()
                cycle.append(g)
            graph_cycle = cycle
        except Exception as exc:
            print(f"graph cycle capture failed ({exc}); eager", file=sys.stderr)

    def run_step(i: int):
        if graph_cycle is not None:
            return graph_cycle[i % len(graph_cycle)].replay()
        # each rank walks a different offset sequence (its DP shard)
        s = ((i + 3 * rank) * B + rank * 17) % max(n - B, 1)
        return step(X[s : s + B], y[s : s + B])

    # ---- warmup ----
    for i in range(args.warmup):
        run_step(i)
    if on_gpu:
        torch.cuda.synchronize()
    if dist_ctx is not None:
        dist_ctx.barrier()
    if on_gpu:
        torch.cuda.synchronize()   # NCCL barrier is stream-ordered: make sure
                                   # it has RUN before any rank reads the clock

    # ---- timed region: exactly K steps ----
    t0 = time.perf_counter()
    for i in range(args.steps):
        run_step(args.warmup + i)
    if on_gpu:
        torch.cuda.synchronize()
    if dist_ctx is not None:
        dist_ctx.barrier()
    if on_gpu:
        torch.cuda.synchronize()
    elapsed = time.perf_counter() - t0

    # max over ranks (slowest rank defines the job)
    if dist_ctx is not None:
        elapsed = dist_ctx.all_reduce_scalar(elapsed, op="max")

    ms_per_step = elapsed / args.steps * 1000.0
    samples_per_sec = (B * world * args.steps) / elapsed

    # ---- accuracy probe (outside the timed region, rank 0, 1-process GPU) --
    accuracy = None
    run_probe = args.accuracy == "on" or (
        args.accuracy == "auto" and on_gpu and world == 1)
    if run_probe and rank == 0:
        accuracy = accuracy_probe(args.accuracy_epochs, device)

    if rank == 0:
        n_gpus = world if on_gpu else args.gpus
        print(json.dumps({
            "metric": f"training samples/sec, {args.endpoints}-endpoint "
                      "3-resource estimation model",
            "value": round(samples_per_sec, 2),
            "unit": "windows/s",
            "n_gpus": n_gpus,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(ms_per_step, 3),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": args.dtype if on_gpu else "fp32",
            "data": "synthetic",
            "accuracy": accuracy,
            "config": {
                "model": "deeprest-amd (attention encoder + call-graph + GRU decoders)",
                "endpoints": args.endpoints,
                "components": args.components,
                "num_metrics": spec.num_metrics,
                "num_paths": spec.num_paths,
                "global_batch": B * world,
                "seq_len": T,
                "parallelism": f"dp{world}",
            },
        }))
    return 0


if __name__ == "__main__":
    sys.exit(main())
