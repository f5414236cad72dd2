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
