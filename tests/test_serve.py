import numpy as np
import pytest
import torch

from deeprest_amd.data.synthetic import SyntheticApp, SyntheticAppConfig
from deeprest_amd.data.synthesizer import TraceSynthesizer
from deeprest_amd.engine.config import DataConfig, EngineConfig, TrainConfig
from deeprest_amd.engine.trainer import Trainer
from deeprest_amd.models.net import DeepRestNetConfig
from deeprest_amd.serve.anomaly import AnomalyScorer
from deeprest_amd.serve.predictor import Predictor
from deeprest_amd.serve.results import ResultsStore, build_results_entry


@pytest.fixture(scope="module")
def trained(tmp_path_factory):
    # module-scoped fixtures instantiate BEFORE the function-scoped autouse
    # seed fixture — seed here or the trained model depends on test order
    torch.manual_seed(0)
    np.random.seed(0)
    tmp = tmp_path_factory.mktemp("serve")
    app = SyntheticApp(SyntheticAppConfig(
        n_apis=4, n_components=5, windows_per_day=60, n_days=2, seed=33))
    raw = app.generate_raw()
    data = app.generate_featurized()
    cfg = EngineConfig()
    cfg.data = DataConfig(step_size=20, split=0.4)
    cfg.train = TrainConfig(epochs=1, batch_size=8, run_baselines=False,
                            log_every=0, checkpoint_path=str(tmp / "ckpt.pt"))
    cfg.model = DeepRestNetConfig(d_model=32, n_heads=4, n_layers=1, d_ff=64,
                                  hidden=16, comp_dim=8, dropout=0.0)
    trainer = Trainer(data, cfg, device=torch.device("cpu"))
    trainer.train()
    return app, raw, data, cfg, str(tmp / "ckpt.pt")


def test_predictor_from_checkpoint(trained):
    app, raw, data, cfg, ckpt = trained
    pred = Predictor.from_checkpoint(ckpt, device=torch.device("cpu"))
    windows = np.asarray(data.traffic[:25], dtype=np.float64)[None].repeat(2, axis=0)
    # use proper (N, T, P) windows
    from deeprest_amd.data.windows import sliding_window

    w = sliding_window(np.asarray(data.traffic, dtype=np.float64), 20)[:3]
    out = pred.predict(w)
    assert set(out.keys()) == set(data.metric_names)
    for v in out.values():
        assert v.shape == (3, 20, 3)
        assert (v >= 0).all()


def test_predictor_what_if(trained):
    app, raw, data, cfg, ckpt = trained
    pred = Predictor.from_checkpoint(ckpt, device=torch.device("cpu"))
    syn = TraceSynthesizer(feature_space=app.feature_space).fit(raw)
    plan = [{app.apis[0]: 5, app.apis[1]: 2}] * 30
    out = pred.predict_what_if(syn, plan, step_size=20,
                               rng=np.random.default_rng(0))
    assert set(out.keys()) == set(data.metric_names)


def test_results_store_schema(tmp_path):
    measurement = np.abs(np.random.default_rng(0).normal(50, 10, size=300))
    preds = {
        est: np.abs(np.random.default_rng(1).normal(50, 10, size=(2, 60)))
        for est in ("bl-resrc", "bl-api", "bl-trace", "ours")
    }
    entry = build_results_entry(measurement, preds, calls=[np.arange(300)],
                                train_len=180)
    # exact key layout the web-demo reader expects (dataloader.py:112-125)
    for est in ("bl-resrc", "bl-api", "bl-trace", "ours"):
        assert f"prediction_{est}" in entry
        assert f"scale_{est}" in entry
        assert len(entry[f"scale_{est}"]) == 2
        assert len(entry[f"prediction_{est}"]) == 120
    assert "scale_groundtruth" in entry
    assert len(entry["measurement"]) == 300

    store = ResultsStore()
    store.add("exp1-waves_seen-1x", "frontend", "cpu", entry)
    p = str(tmp_path / "results.pkl")
    store.save(p)
    loaded = ResultsStore.load(p)
    assert loaded.get("exp1-waves_seen-1x", "frontend", "cpu")["scale_ours"] == entry["scale_ours"]


def test_rest_results_browsing(tmp_path):
    """GET /results mirrors the web-demo DataLoader surface over results.pkl."""
    from starlette.testclient import TestClient

    from deeprest_amd.serve.api import create_app

    measurement = np.abs(np.random.default_rng(0).normal(50, 10, size=300))
    preds = {est: np.abs(np.random.default_rng(1).normal(50, 10, size=(2, 60)))
             for est in ("bl-resrc", "bl-api", "bl-trace", "ours")}
    entry = build_results_entry(measurement, preds, calls=[np.arange(300)],
                                train_len=180)
    store = ResultsStore()
    store.add("exp1-waves_seen-1x", "frontend", "cpu", entry)
    p = str(tmp_path / "results.pkl")
    store.save(p)

    client = TestClient(create_app(results_path=p))
    # the results-browser page serves alongside the JSON surface
    page = client.get("/demo/results")
    assert page.status_code == 200 and "results browser" in page.text
    assert client.get("/results").json() == {"experiments": ["exp1-waves_seen-1x"]}
    assert client.get("/results/exp1-waves_seen-1x").json() == {"frontend": ["cpu"]}
    got = client.get("/results/exp1-waves_seen-1x/frontend/cpu").json()
    assert "prediction_ours" in got and "scale_groundtruth" in got
    assert len(got["measurement"]) == 300
    assert client.get("/results/nope").status_code == 404
    # without a results store the endpoints answer 400, not crash
    bare = TestClient(create_app())
    assert bare.get("/results").status_code == 400


def test_anomaly_scorer_flags_injected_cpu_thief():
    T = 100
    rng = np.random.default_rng(2)
    q50 = 50 + 5 * np.sin(np.arange(T) / 7)
    q05, q95 = q50 - 10, q50 + 10
    measured = q50 + rng.normal(0, 2, T)
    measured[40:55] += 60.0  # cryptojacking burst not justified by traffic
    rep = AnomalyScorer(threshold=0.25, min_run=3).score(measured, q05, q50, q95,
                                                         metric="svc_cpu")
    assert rep.is_anomalous
    assert any(s <= 40 < e or (s >= 40 and e <= 56) for s, e in rep.windows)
    # clean region unflagged
    assert not rep.flags[:35].any()
    assert not rep.flags[60:].any()


def test_anomaly_scorer_clean_series_unflagged():
    T = 80
    q50 = np.full(T, 30.0)
    rep = AnomalyScorer().score(q50 + 0.5, q50 - 5, q50, q50 + 5)
    assert not rep.is_anomalous


def test_rest_api_end_to_end(trained):
    from starlette.testclient import TestClient

    from deeprest_amd.serve.api import create_app

    app_obj, raw, data, cfg, ckpt = trained
    pred = Predictor.from_checkpoint(ckpt, device=torch.device("cpu"))
    api = create_app(predictor=pred)
    client = TestClient(api)

    r = client.get("/health")
    assert r.status_code == 200 and r.json()["status"] == "ok"

    r = client.post("/ingest", json=raw[:40])
    assert r.status_code == 200 and r.json()["windows_total"] == 40

    r = client.post("/ingest", json=[{"bad": "window"}])
    assert r.status_code == 422

    r = client.post("/featurize")
    assert r.status_code == 200
    assert r.json()["num_windows"] == 40

    r = client.get("/apis")
    assert r.status_code == 200
    apis = r.json()["apis"]
    assert len(apis) == 4

    r = client.post("/estimate", json={"traffic_plan": [{apis[0]: 3}] * 25,
                                       "step_size": 20, "seed": 0})
    assert r.status_code == 200
    body = r.json()
    assert body["quantiles"] == [0.05, 0.50, 0.95]
    assert set(body["predictions"].keys()) == set(data.metric_names)

    measured = {data.metric_names[0]: [100.0] * 10}
    predicted = {data.metric_names[0]: [[10.0, 20.0, 30.0]] * 10}
    r = client.post("/anomaly", json={"measured": measured, "predicted": predicted})
    assert r.status_code == 200
    assert r.json()[data.metric_names[0]]["anomalous"] is True


def test_demo_page_served(trained):
    from starlette.testclient import TestClient

    from deeprest_amd.serve.api import create_app

    client = TestClient(create_app())
    r = client.get("/demo")
    assert r.status_code == 200
    assert "deeprest-amd" in r.text and "canvas" in r.text


def test_estimate_uses_model_feature_space(trained):
    """A checkpoint-loaded predictor carries its frozen call-path space;
    /featurize must featurize INTO it so /estimate's synthesized vectors
    match the model's input width even when the ingested windows only cover
    a subset of paths (regression: this used to 500 with a shape mismatch)."""
    from starlette.testclient import TestClient

    from deeprest_amd.serve.api import create_app

    app_obj, raw, data, cfg, ckpt = trained
    pred = Predictor.from_checkpoint(ckpt, device=torch.device("cpu"))
    client = TestClient(create_app(predictor=pred))
    # ingest only a few windows with truncated trace lists: a strict subset
    subset = [{"metrics": w["metrics"], "traces": w["traces"][:1]} for w in raw[:10]]
    assert client.post("/ingest", json=subset).status_code == 200
    r = client.post("/featurize")
    assert r.status_code == 200
    if pred.feature_space is not None:
        assert r.json()["frozen_to_model_space"] is True
        assert r.json()["num_paths"] == pred.model.spec.num_paths
    apis = client.get("/apis").json()["apis"]
    if apis:
        out = client.post("/estimate", json={
            "traffic_plan": [{apis[0]: 2} for _ in range(25)],
            "step_size": 20, "seed": 0})
        # either a clean answer or a clean 422 (never a 500 shape crash)
        assert out.status_code in (200, 422)
        if out.status_code == 200:
            assert out.json()["predictions"]


def test_cryptojacking_detection_end_to_end(trained):
    """The reference's sanity-check story: inject a traffic-unjustified CPU
    burst into the measured series and detect it against the TRAINED model's
    own quantile band (reference locust/pow.py + README.md:3)."""
    from deeprest_amd.data.windows import sliding_window

    app_obj, raw, data, cfg, ckpt = trained
    pred = Predictor.from_checkpoint(ckpt, device=torch.device("cpu"))
    W = 20
    windows = sliding_window(np.asarray(data.traffic, dtype=np.float64), W)[:1]
    out = pred.predict(windows)                      # {metric: (1, W, 3)}
    metric = data.metric_names[0]
    q = out[metric][0]                               # (W, 3)
    measured = q[:, 1].copy()                        # sits on the median
    # the thief's burst must dwarf the model's own uncertainty band (which
    # an undertrained model can make arbitrarily wide)
    burst = 50.0 * float(np.mean(q[:, 2] - q[:, 0])) + 50.0
    measured[8:14] += burst

    scorer = AnomalyScorer(threshold=1.0, min_run=3)
    rep = scorer.score(measured, q[:, 0], q[:, 1], q[:, 2], metric=metric)
    assert rep.is_anomalous
    assert any(s >= 7 and e <= 15 for s, e in rep.windows)
    # the untouched region stays quiet (measured == predicted median)
    assert not rep.flags[:7].any() and not rep.flags[15:].any()


def test_results_entry_reanchors_monotone_metrics():
    """Memory/usage scale factors re-anchor each query window to the last
    learning-period value (reference: web-demo/dataloader.py:143-156) —
    without it a monotonically growing metric shows absurd scales."""
    from deeprest_amd.serve.results import build_results_entry

    # monotone growth: learning period ends at 100, query continues to 160
    meas = np.arange(1.0, 161.0)          # 160 steps
    t_train = 100
    pred = np.array([[130.0, 135.0, 140.0]])   # one query window, offset high
    plain = build_results_entry(meas, {"ours": pred}, train_len=t_train)
    re = build_results_entry(meas, {"ours": pred}, train_len=t_train,
                             reanchor=True)
    train_peak = 100.0
    # plain: raw peak / train peak
    assert plain["scale_ours"][0] == 140.0 / train_peak
    # re-anchored: window shifted to start at meas[99] = 100 -> peak 110
    assert re["scale_ours"][0] == (140.0 - 130.0 + 100.0) / train_peak
    # ground-truth scale re-anchors the query segment the same way
    seg = meas[100:103]
    expected_gt = (seg[-1] - seg[0] + 100.0) / train_peak
    assert abs(re["scale_groundtruth"][0] - expected_gt) < 1e-12
    # separate-timeline form: anchor passed explicitly
    re2 = build_results_entry(meas[100:], {"ours": pred}, train_len=0,
                              train_peak=train_peak, reanchor=True,
                              anchor_value=100.0)
    assert re2["scale_ours"][0] == (140.0 - 130.0 + 100.0) / train_peak


def test_old_checkpoint_with_biases_loads(tmp_path):
    """Checkpoints from before the biasless default carried biases and no
    linear_bias key — the from_full_state shim must restore them."""
    from deeprest_amd.data.synthetic import SyntheticApp, SyntheticAppConfig
    from deeprest_amd.models.net import (DeepRestNet, DeepRestNetConfig,
                                         build_model_spec)

    app = SyntheticApp(SyntheticAppConfig(
        n_apis=3, n_components=4, windows_per_day=40, n_days=1, seed=3))
    data = app.generate_featurized()
    spec = build_model_spec(data)
    old = DeepRestNet(spec, DeepRestNetConfig(
        d_model=32, n_heads=2, n_layers=1, d_ff=64, hidden=16, comp_dim=8,
        linear_bias=True))
    state = old.full_state()
    del state["config"]["linear_bias"]          # emulate the old format
    loaded = DeepRestNet.from_full_state(state)
    assert loaded.cfg.linear_bias is True
    assert loaded.in_proj.bias is not None
    x = torch.rand(2, 10, spec.num_paths)
    old.eval(), loaded.eval()          # dropout off for the equality check
    torch.testing.assert_close(loaded(x), old(x))


def test_predictor_restores_target_transform(trained, tmp_path):
    """log1p-trained checkpoints serve with expm1 denormalization."""
    import numpy as np

    from deeprest_amd.engine.config import EngineConfig, DataConfig, TrainConfig
    from deeprest_amd.engine.trainer import Trainer
    from deeprest_amd.models.net import DeepRestNetConfig
    from deeprest_amd.data.synthetic import SyntheticApp, SyntheticAppConfig

    app = SyntheticApp(SyntheticAppConfig(
        n_apis=3, n_components=4, windows_per_day=60, n_days=2, seed=3))
    data = app.generate_featurized()
    cfg = EngineConfig(
        data=DataConfig(step_size=20, split=0.4, target_transform="log1p"),
        train=TrainConfig(epochs=1, batch_size=16, run_baselines=False,
                          log_every=0),
        model=DeepRestNetConfig(d_model=32, n_heads=2, n_layers=1, d_ff=64,
                                hidden=16, comp_dim=8, dropout=0.0))
    tr = Trainer(data, cfg, device=torch.device("cpu"))
    tr.train()
    ckpt = str(tmp_path / "log1p.pt")
    tr.save(ckpt, 1)
    pred = Predictor.from_checkpoint(ckpt, device=torch.device("cpu"))
    assert pred.target_transform == "log1p"
    from deeprest_amd.data.windows import sliding_window

    w = sliding_window(np.asarray(data.traffic, dtype=np.float64), 20)[:4]
    out = pred.predict(w)
    # denormalized predictions land in raw-resource magnitude, not log space
    first = out[data.metric_names[0]]
    raw_med = float(np.median(data.resources[data.metric_names[0]]))
    assert float(np.median(first)) > np.log1p(raw_med)  # clearly not log-space


def test_predict_endpoint_with_micro_batching(trained):
    from starlette.testclient import TestClient

    from deeprest_amd.serve.api import create_app

    app_syn, raw, data, cfg, ckpt = trained
    client = TestClient(create_app(checkpoint_path=ckpt, micro_batch=True))
    h = client.get("/health").json()
    assert h["model_loaded"] is True
    P = len(data.feature_space)
    w = np.asarray(data.traffic[:24], dtype=float).reshape(2, 12, P)
    r = client.post("/predict", json={"windows": w.tolist()})
    assert r.status_code == 200, r.text
    body = r.json()
    assert body["quantiles"] == [0.05, 0.5, 0.95]
    m0 = data.metric_names[0]
    arr = np.asarray(body["predictions"][m0])
    assert arr.shape == (2, 12, 3)
    assert body["micro_batch"]["requests_served"] >= 1
    # bad shape rejected
    r2 = client.post("/predict", json={"windows": [[1, 2], [3, 4]]})
    assert r2.status_code == 422
