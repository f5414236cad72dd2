import subprocess
import sys


from deeprest_amd.data.contract import save_raw_data
from deeprest_amd.data.synthetic import SyntheticApp, SyntheticAppConfig


def test_cli_featurize_and_synthesize(tmp_path):
    app = SyntheticApp(SyntheticAppConfig(
        n_apis=3, n_components=4, windows_per_day=20, n_days=1, seed=2))
    raw_p = str(tmp_path / "raw.pkl")
    save_raw_data(app.generate_raw(), raw_p)

    out_p = str(tmp_path / "input.pkl")
    r = subprocess.run(
        [sys.executable, "-m", "deeprest_amd.cli", "featurize",
         "--raw", raw_p, "--out", out_p],
        capture_output=True, text=True, timeout=300,
    )
    assert r.returncode == 0, r.stderr
    assert "featurized 20 windows" in r.stdout

    r = subprocess.run(
        [sys.executable, "-m", "deeprest_amd.cli", "synthesize", "--raw", raw_p],
        capture_output=True, text=True, timeout=300,
    )
    assert r.returncode == 0, r.stderr
    assert "3 API endpoints are found" in r.stdout or "3 API endpoints found" in r.stdout


def test_cli_full_pipeline(tmp_path):
    """featurize -> train(checkpoint) -> serve app wired from the artifacts:
    one end-to-end pass across module boundaries."""
    import numpy as np
    from starlette.testclient import TestClient

    from deeprest_amd.cli import main
    from deeprest_amd.data.contract import save_raw_data
    from deeprest_amd.data.synthetic import SyntheticApp, SyntheticAppConfig
    from deeprest_amd.serve.api import create_app

    app = SyntheticApp(SyntheticAppConfig(
        n_apis=4, n_components=5, windows_per_day=150, n_days=1, seed=21))
    # scale traffic way down: full-size plans emit tens of thousands of span
    # trees and make this an expensive test for no extra coverage
    raw = app.generate_raw(plan=app.traffic_plan(scale=0.03))
    raw_p = str(tmp_path / "raw.pkl")
    save_raw_data(raw, raw_p)

    inp = str(tmp_path / "input.pkl")
    assert main(["featurize", "--raw", raw_p, "--out", inp]) == 0

    ckpt = str(tmp_path / "ckpt.pt")
    assert main([
        "train",
        "--set", f"data.input_path={inp}",
        "--set", "data.step_size=20",
        "--set", "data.split=0.5",
        "--set", "train.epochs=1",
        "--set", "train.run_baselines=false",
        "--set", "train.log_every=0",
        "--set", f"train.checkpoint_path={ckpt}",
        "--set", "model.d_model=32", "--set", "model.n_heads=4",
        "--set", "model.n_layers=1", "--set", "model.d_ff=64",
        "--set", "model.hidden=16", "--set", "model.comp_dim=8",
        "--set", "model.dropout=0.0",
    ]) == 0

    client = TestClient(create_app(checkpoint_path=ckpt))
    assert client.get("/health").json()["model_loaded"] is True
    # ingest the raw windows over REST, featurize, and run a what-if estimate
    payload = [{"metrics": w["metrics"], "traces": w["traces"][:5]} for w in raw[:30]]
    assert client.post("/ingest", json=payload).status_code == 200
    assert client.post("/featurize").status_code == 200
    apis = client.get("/apis").json()["apis"]
    assert apis
    r = client.post("/estimate", json={
        "traffic_plan": [{apis[0]: 3} for _ in range(20)],
        "step_size": 20, "seed": 1})
    assert r.status_code == 200
    preds = r.json()["predictions"]
    assert preds
    for series in preds.values():
        assert np.isfinite(np.asarray(series, dtype=np.float64)).all()
