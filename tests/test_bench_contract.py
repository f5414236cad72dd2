"""bench.py driver-contract test.

The round-end driver parses exactly one JSON line from rank 0 with a fixed
key set (see the repo instructions); this test locks that contract on the
CPU path so kernel/bench edits can't silently break it.
"""

import json
import os
import subprocess
import sys

import numpy as np

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_bench_json_contract():
    out = subprocess.run(
        [sys.executable, "bench.py", "--steps", "2", "--warmup", "1",
         "--batch", "8", "--endpoints", "6", "--components", "5",
         "--seq-len", "12"],
        cwd=REPO, capture_output=True, text=True, timeout=600,
    )
    assert out.returncode == 0, out.stderr[-2000:]
    lines = [l for l in out.stdout.strip().splitlines() if l.startswith("{")]
    assert len(lines) == 1, f"expected exactly one JSON line, got: {out.stdout!r}"
    d = json.loads(lines[0])

    for key in ("metric", "value", "unit", "n_gpus", "steps", "warmup",
                "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
                "dtype", "data", "config"):
        assert key in d, f"missing contract key {key}"
    assert d["value"] > 0 and d["ms_per_step"] > 0
    assert d["steps"] == 2 and d["warmup"] == 1
    assert d["higher_is_better"] is True
    assert d["scaling"] == "weak"
    assert d["data"] == "synthetic"
    assert d["vs_baseline"] is None  # BASELINE.json publishes no number
    cfg = d["config"]
    for key in ("model", "global_batch", "seq_len", "parallelism"):
        assert key in cfg
    assert cfg["global_batch"] == 8
    assert cfg["parallelism"] == "dp1"


def test_bench_torchrun_dp2_contract():
    """The driver's N>1 launch path: torchrun --nnodes=1 --nproc-per-node 2
    with MASTER_ADDR=127.0.0.1 (gloo on CPU here, RCCL on the GPU node)."""
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29537", "bench.py", "--gpus", "2",
         "--steps", "2", "--warmup", "1", "--batch", "8",
         "--endpoints", "6", "--components", "5", "--seq-len", "12"],
        cwd=REPO, capture_output=True, text=True, timeout=420,
    )
    assert out.returncode == 0, out.stderr[-2000:]
    lines = [l for l in out.stdout.strip().splitlines() if l.startswith("{")]
    assert len(lines) == 1, f"exactly one JSON line from rank 0, got {lines}"
    d = json.loads(lines[0])
    assert d["n_gpus"] == 2
    assert d["config"]["parallelism"] == "dp2"
    assert d["config"]["global_batch"] == 16  # whole-job aggregate


def test_bench_accuracy_key_default_off_on_cpu():
    """CPU default run: the accuracy key is present (contract) but null —
    the probe only auto-runs on single-process GPU benches."""
    out = subprocess.run(
        [sys.executable, "bench.py", "--steps", "1", "--warmup", "0",
         "--batch", "8", "--endpoints", "6", "--components", "5",
         "--seq-len", "12"],
        cwd=REPO, capture_output=True, text=True, timeout=600,
    )
    assert out.returncode == 0, out.stderr[-2000:]
    d = json.loads([l for l in out.stdout.splitlines() if l.startswith("{")][0])
    assert "accuracy" in d and d["accuracy"] is None


def test_accuracy_probe_shape():
    """The MAE probe emits the three-estimator aggregate the driver line
    carries (tiny config so it runs on CPU in seconds)."""
    sys.path.insert(0, REPO)
    import torch
    from bench import accuracy_probe

    acc = accuracy_probe(1, torch.device("cpu"), n_apis=3, n_components=4,
                         windows_per_day=40, n_days=2, step_size=10,
                         baseline_epochs=2)
    assert set(acc["mean_median_abs_err"]) == {"resrc", "comp", "deepr"}
    assert acc["metrics"] == 5 * 3          # (4 comps + frontend) x 3 resources
    assert 0 <= acc["deepr_beats_comp"] <= acc["metrics"]
    assert acc["probe_seconds"] > 0
    # the extended-convergence row the driver line carries
    assert acc["extended"]["epochs"] == 2
    assert np.isfinite(acc["extended"]["deepr"])


def test_bench_torchrun_dp8_contract():
    """The driver's largest launch shape: 8 ranks (gloo on CPU here, RCCL
    on the node). Exercises 8-reader dataset sharing, rank walks, and the
    rank-0-only JSON line at world size 8."""
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "8", "--master-addr", "127.0.0.1",
         "--master-port", "29538", "bench.py", "--gpus", "8",
         "--steps", "1", "--warmup", "0", "--batch", "4",
         "--endpoints", "6", "--components", "5", "--seq-len", "12"],
        cwd=REPO, capture_output=True, text=True, timeout=600,
    )
    assert out.returncode == 0, out.stderr[-2000:]
    lines = [l for l in out.stdout.strip().splitlines() if l.startswith("{")]
    assert len(lines) == 1, f"exactly one JSON line from rank 0, got {lines}"
    d = json.loads(lines[0])
    assert d["n_gpus"] == 8
    assert d["config"]["parallelism"] == "dp8"
    assert d["config"]["global_batch"] == 32
