import numpy as np

from deeprest_amd.data.collector import Collector, span_tree_from_jaeger
from deeprest_amd.data.contract import validate_raw_data
from deeprest_amd.data.featurize import Featurizer


def make_jaeger_trace(trace_id, t0_us, svc_ops):
    """svc_ops: list of (spanID, service, op, parentID or None)."""
    spans = []
    for sid, svc, op, parent in svc_ops:
        span = {
            "traceID": trace_id,
            "spanID": sid,
            "operationName": op,
            "startTime": t0_us,
            "serviceName": svc,
            "references": [],
        }
        if parent is not None:
            span["references"].append({"refType": "CHILD_OF", "spanID": parent})
        spans.append(span)
    return {"traceID": trace_id, "spans": spans}


def test_span_tree_reconstruction():
    tr = make_jaeger_trace("t1", 0, [
        ("a", "nginx", "/compose", None),
        ("b", "text-svc", "parse", "a"),
        ("c", "user-db", "find", "b"),
        ("d", "media-svc", "upload", "a"),
    ])
    tree = span_tree_from_jaeger(tr)
    assert tree["component"] == "nginx"
    assert len(tree["children"]) == 2
    kids = {c["component"] for c in tree["children"]}
    assert kids == {"text-svc", "media-svc"}
    text = next(c for c in tree["children"] if c["component"] == "text-svc")
    assert text["children"][0]["component"] == "user-db"


def test_collector_windows_and_contract():
    col = Collector(window_sec=5.0, t0=0.0)
    # traces in windows 0 and 2
    col.add_traces([
        make_jaeger_trace("t1", int(1e6), [("a", "nginx", "/x", None)]),
        make_jaeger_trace("t2", int(11e6), [("b", "nginx", "/y", None)]),
    ])
    col.add_metric_samples([
        {"component": "nginx", "resource": "cpu", "value": 5.0, "timestamp": 1.0},
        {"component": "nginx", "resource": "cpu", "value": 7.0, "timestamp": 6.0},
        {"component": "db", "resource": "memory", "value": 100.0, "timestamp": 11.0},
    ])
    windows = col.windows()
    assert len(windows) == 3
    validate_raw_data(windows)
    # windows feed the featurizer directly
    data = Featurizer(use_native=False).fit_transform(windows)
    assert data.num_windows == 3
    assert data.num_paths == 2
    # gauge carry-forward: window 1 keeps cpu=7, memory still 0
    cpu = data.resources["nginx_cpu"]
    np.testing.assert_allclose(cpu, [5.0, 7.0, 7.0])
    mem = data.resources["db_memory"]
    np.testing.assert_allclose(mem, [0.0, 0.0, 100.0])
    # trace placement
    assert data.invocations["general"].tolist() == [1, 0, 1]


def test_collector_jaeger_process_map():
    tr = {
        "traceID": "t",
        "processes": {"p1": {"serviceName": "svc-via-process"}},
        "spans": [{
            "traceID": "t", "spanID": "s1", "operationName": "op",
            "startTime": 0, "processID": "p1", "references": [],
        }],
    }
    tree = span_tree_from_jaeger(tr)
    assert tree["component"] == "svc-via-process"


def test_span_tree_child_listed_before_parent():
    """Two-pass reconstruction is order-independent (Jaeger export order is
    arbitrary)."""
    trace = {
        "spans": [
            {"spanID": "c", "operationName": "leaf", "serviceName": "svc-b",
             "references": [{"refType": "CHILD_OF", "spanID": "r"}],
             "startTime": 2_000_000},
            {"spanID": "r", "operationName": "root", "serviceName": "svc-a",
             "references": [], "startTime": 1_000_000},
        ]
    }
    tree = span_tree_from_jaeger(trace)
    assert tree["component"] == "svc-a"
    assert tree["children"][0]["component"] == "svc-b"
    assert tree["children"][0]["operation"] == "leaf"


def test_span_tree_orphan_reference_becomes_root():
    """A span whose parent never arrived is treated as the root (partial
    traces happen under sampling/drops); extra orphans are dropped."""
    trace = {
        "spans": [
            {"spanID": "x", "operationName": "op", "serviceName": "svc",
             "references": [{"refType": "CHILD_OF", "spanID": "missing"}],
             "startTime": 0},
        ]
    }
    tree = span_tree_from_jaeger(trace)
    assert tree is not None and tree["component"] == "svc"


def test_collector_empty_returns_no_windows():
    assert Collector().windows() == []


def test_jaeger_to_trained_model_end_to_end():
    """Full seam: Jaeger-style exports + Prometheus-style samples ->
    collector windows -> featurizer -> a training step of the real model.
    Catches drift between the collection plane's output and what the
    engine's data path accepts."""
    import torch

    from deeprest_amd.engine.config import DataConfig, EngineConfig, TrainConfig
    from deeprest_amd.engine.trainer import Trainer
    from deeprest_amd.models.net import DeepRestNetConfig

    rng = np.random.default_rng(0)
    col = Collector(window_sec=5.0, t0=0.0)
    # 120 windows of alternating request shapes + drifting cpu samples
    for w in range(120):
        t_us = int((w * 5.0 + 0.5) * 1e6)
        n_calls = 1 + int(2 + 2 * np.sin(w / 10) + rng.integers(0, 2))
        for c in range(n_calls):
            shape = [("root", "nginx", "/compose", None),
                     ("s1", "text-svc", "/parse", "root")]
            if (w + c) % 2 == 0:
                shape.append(("s2", "user-db", "/find", "s1"))
            col.add_traces([make_jaeger_trace(f"w{w}c{c}", t_us, shape)])
        load = n_calls * 10.0 + rng.normal(0, 0.5)
        col.add_metric_samples([
            {"component": "nginx", "resource": "cpu", "value": load,
             "timestamp": w * 5.0 + 1.0},
            {"component": "user-db", "resource": "cpu", "value": load * 0.5,
             "timestamp": w * 5.0 + 1.0},
        ])
    windows = col.windows()
    validate_raw_data(windows)
    data = Featurizer(use_native=False).fit_transform(windows)
    assert data.num_windows >= 120

    cfg = EngineConfig()
    cfg.data = DataConfig(step_size=20, split=0.5)
    cfg.train = TrainConfig(epochs=1, batch_size=8, run_baselines=False,
                            log_every=0)
    cfg.model = DeepRestNetConfig(d_model=32, n_heads=4, n_layers=1, d_ff=64,
                                  hidden=16, comp_dim=8, dropout=0.0)
    res = Trainer(data, cfg, device=torch.device("cpu")).train()
    assert np.isfinite(res.train_losses).all()
