import numpy as np
import torch

from deeprest_amd.data.synthetic import SyntheticApp, SyntheticAppConfig
from deeprest_amd.engine.config import DataConfig, EngineConfig, TrainConfig
from deeprest_amd.engine.experiment import run_experiment
from deeprest_amd.models.net import DeepRestNetConfig
from deeprest_amd.serve.results import ResultsStore


def test_run_experiment_produces_results_schema(tmp_path):
    app = SyntheticApp(SyntheticAppConfig(
        n_apis=3, n_components=4, windows_per_day=80, n_days=2, seed=55))
    data = app.generate_featurized()
    cfg = EngineConfig()
    cfg.data = DataConfig(step_size=20, split=0.4)
    cfg.train = TrainConfig(epochs=1, batch_size=8, baseline_epochs=2,
                            eval_cycles=3, log_every=0)
    cfg.model = DeepRestNetConfig(d_model=32, n_heads=4, n_layers=1, d_ff=64,
                                  hidden=16, comp_dim=8, dropout=0.0)

    calls = [app.traffic_plan()[:, i] for i in range(3)]
    store = run_experiment(data, "synthetic-waves_seen-1x", config=cfg,
                           device=torch.device("cpu"), calls_series=calls)
    assert store.experiments() == ["synthetic-waves_seen-1x"]
    # every (component, resource) of the app appears
    comps = store.results["synthetic-waves_seen-1x"]
    assert set(comps.keys()) == set(app.all_components)
    entry = comps[app.all_components[0]]["cpu"]
    for est in ("bl-resrc", "bl-api", "bl-trace", "ours"):
        assert f"prediction_{est}" in entry
        assert f"scale_{est}" in entry
        assert all(np.isfinite(entry[f"prediction_{est}"]))
    assert len(entry["calls"]) == 3
    assert "scale_groundtruth" in entry

    # persists in the reference's on-disk format
    p = str(tmp_path / "results.pkl")
    store.save(p)
    loaded = ResultsStore.load(p)
    assert loaded.get("synthetic-waves_seen-1x", app.all_components[0], "cpu")


def test_scenario_suite_unseen_traffic():
    """Train once on normal traffic, query under unseen scale/shape/
    composition — the reference's headline evaluation axes (locustfile
    variants); one results.pkl experiment per scenario."""
    import numpy as np

    from deeprest_amd.data.synthetic import SyntheticApp, SyntheticAppConfig
    from deeprest_amd.engine.config import DataConfig, EngineConfig, TrainConfig
    from deeprest_amd.engine.experiment import run_scenario_suite
    from deeprest_amd.models.net import DeepRestNetConfig

    app = SyntheticApp(SyntheticAppConfig(
        n_apis=6, n_components=6, windows_per_day=200, n_days=1, seed=9))
    cfg = EngineConfig()
    cfg.data = DataConfig(step_size=20, split=0.5)
    cfg.train = TrainConfig(epochs=2, batch_size=16, baseline_epochs=3,
                            eval_cycles=3, run_baselines=True, log_every=0)
    cfg.model = DeepRestNetConfig(d_model=32, n_heads=4, n_layers=1, d_ff=64,
                                  hidden=16, comp_dim=8, dropout=0.0)
    store = run_scenario_suite(app, base_name="t", config=cfg,
                               device=__import__("torch").device("cpu"))
    exps = store.experiments()
    assert sorted(exps) == sorted([
        "t-waves_seen-1x", "t-waves_unseen-3x", "t-flat_unseen-1x",
        "t-waves_unseen_compositions-1x",
    ])
    for exp in exps:
        comps = store.results[exp]
        assert comps, exp
        some = next(iter(comps.values()))
        entry = next(iter(some.values()))
        for est in ("bl-resrc", "bl-api", "bl-trace", "ours"):
            pred = np.asarray(entry[f"prediction_{est}"])
            assert np.isfinite(pred).all()
            assert (pred >= 0).all()
        assert np.isfinite(entry["measurement"]).all()
    # the 3x-scale scenario's ground truth really is bigger traffic
    m_seen = np.mean(next(iter(next(iter(
        store.results["t-waves_seen-1x"].values())).values()))["measurement"])
    m_3x = np.mean(next(iter(next(iter(
        store.results["t-waves_unseen-3x"].values())).values()))["measurement"])
    assert m_3x > m_seen


def test_scenario_error_tables_shape():
    import numpy as np

    from deeprest_amd.data.synthetic import SyntheticApp, SyntheticAppConfig
    from deeprest_amd.engine.config import DataConfig, EngineConfig, TrainConfig
    from deeprest_amd.engine.experiment import (run_scenario_suite,
                                                scenario_error_tables)
    from deeprest_amd.models.net import DeepRestNetConfig
    import torch

    app = SyntheticApp(SyntheticAppConfig(
        n_apis=5, n_components=5, windows_per_day=180, n_days=1, seed=4))
    cfg = EngineConfig()
    cfg.data = DataConfig(step_size=20, split=0.5)
    cfg.train = TrainConfig(epochs=1, batch_size=16, baseline_epochs=2,
                            eval_cycles=2, run_baselines=True, log_every=0)
    cfg.model = DeepRestNetConfig(d_model=32, n_heads=4, n_layers=1, d_ff=64,
                                  hidden=16, comp_dim=8, dropout=0.0)
    store = run_scenario_suite(app, base_name="s", config=cfg,
                               device=torch.device("cpu"))
    tables = scenario_error_tables(store)
    assert set(tables.keys()) == set(store.experiments())
    for exp, per_est in tables.items():
        assert set(per_est.keys()) == {"bl-resrc", "bl-api", "bl-trace", "ours"}
        for est, t in per_est.items():
            assert np.isfinite(t["median"]) and t["median"] >= 0


def test_run_experiment_from_raw_wrapper():
    import torch

    from deeprest_amd.data.synthetic import SyntheticApp, SyntheticAppConfig
    from deeprest_amd.engine.config import DataConfig, EngineConfig, TrainConfig
    from deeprest_amd.engine.experiment import run_experiment_from_raw
    from deeprest_amd.models.net import DeepRestNetConfig

    app = SyntheticApp(SyntheticAppConfig(
        n_apis=3, n_components=4, windows_per_day=120, n_days=1, seed=6))
    raw = app.generate_raw(plan=app.traffic_plan(scale=0.05))
    cfg = EngineConfig(
        data=DataConfig(step_size=20, split=0.5),
        train=TrainConfig(epochs=1, batch_size=8, run_baselines=False,
                          baseline_epochs=2, log_every=0),
        model=DeepRestNetConfig(d_model=32, n_heads=4, n_layers=1, d_ff=64,
                                hidden=16, comp_dim=8, dropout=0.0))
    store = run_experiment_from_raw(raw, "from-raw", config=cfg,
                                    device=torch.device("cpu"))
    assert store.experiments() == ["from-raw"]


def test_scenario_suite_five_resources_smoke():
    """Scenario suite end-to-end with the reference's full resource set;
    usage entries carry re-anchored scale factors."""
    import torch

    from deeprest_amd.data.synthetic import ALL_RESOURCES, SyntheticApp, SyntheticAppConfig
    from deeprest_amd.engine.config import DataConfig, EngineConfig, TrainConfig
    from deeprest_amd.engine.experiment import run_scenario_suite, scenario_error_tables
    from deeprest_amd.models.net import DeepRestNetConfig

    app = SyntheticApp(SyntheticAppConfig(
        n_apis=3, n_components=4, resources=ALL_RESOURCES,
        windows_per_day=60, n_days=2, seed=31))
    cfg = EngineConfig(
        data=DataConfig(step_size=20, split=0.4),
        train=TrainConfig(epochs=1, batch_size=16, baseline_epochs=2,
                          log_every=0),
        model=DeepRestNetConfig(d_model=32, n_heads=4, n_layers=1, d_ff=64,
                                hidden=16, comp_dim=8, dropout=0.0))
    cfg.train.conformal = 0.9
    store = run_scenario_suite(app, config=cfg, device=torch.device("cpu"),
                               scenarios=[("waves_unseen-3x", {"scale": 3.0})])
    exp = store.experiments()[0]
    # coverage-under-shift side table (non-monotone resources only)
    cov = store.scenario_coverage[exp]
    assert set(cov.keys()) == {"cpu", "write-iops", "write-tp"}
    for vals in cov.values():
        assert all(0.0 <= v <= 1.0 for v in vals)
    comps = store.results[exp]
    # every component has entries for all five resource types
    for comp, metrics in comps.items():
        assert set(metrics.keys()) == set(ALL_RESOURCES)
        e = metrics["usage"]
        assert "scale_ours" in e and "scale_groundtruth" in e
        assert all(np.isfinite(e["scale_ours"]))
    tables = scenario_error_tables(store)
    assert set(tables[exp].keys()) == {"bl-resrc", "bl-api", "bl-trace", "ours"}
