"""One narrative test of the full user story on CPU:

synthetic app -> raw-data contract -> featurize -> train (residual head,
checkpointed) -> Predictor -> what-if via the trace synthesizer ->
anomaly scoring of traffic-unjustified utilization.

Each stage also has focused tests elsewhere; this is the end-to-end
spine a reference user would follow.
"""

import numpy as np
import torch

from deeprest_amd.data.contract import validate_raw_data
from deeprest_amd.data.featurize import Featurizer
from deeprest_amd.data.synthesizer import TraceSynthesizer
from deeprest_amd.data.synthetic import SyntheticApp, SyntheticAppConfig
from deeprest_amd.engine.config import DataConfig, EngineConfig, TrainConfig
from deeprest_amd.engine.trainer import Trainer
from deeprest_amd.models.net import DeepRestNetConfig
from deeprest_amd.serve.anomaly import AnomalyScorer
from deeprest_amd.serve.predictor import Predictor


def test_full_user_story(tmp_path):
    app = SyntheticApp(SyntheticAppConfig(
        n_apis=4, n_components=5, windows_per_day=70, n_days=2, seed=11))
    raw = app.generate_raw()
    validate_raw_data(raw)

    data = Featurizer().fit_transform(raw)
    assert data.num_paths > 0 and data.metric_names

    ckpt = str(tmp_path / "e2e.pt")
    cfg = EngineConfig(
        data=DataConfig(step_size=20, split=0.4),
        train=TrainConfig(epochs=2, batch_size=16, run_baselines=False,
                          log_every=0, residual_base="trace-ridge",
                          checkpoint_path=ckpt),
        model=DeepRestNetConfig(d_model=32, n_heads=2, n_layers=1, d_ff=64,
                                hidden=16, comp_dim=8, dropout=0.0))
    trainer = Trainer(data, cfg, device=torch.device("cpu"))
    result = trainer.train()
    assert np.isfinite(result.train_losses).all()

    pred = Predictor.from_checkpoint(ckpt, device=torch.device("cpu"))
    assert pred.feature_space is not None
    assert pred.residual_ridge is not None

    # what-if: synthesize traffic for a hypothetical mix of the two most
    # popular endpoints and predict quantile bands
    syn = TraceSynthesizer().fit(raw)
    plan = [{syn.apis[0]: 10, syn.apis[1]: 4} for _ in range(30)]
    out = pred.predict_what_if(syn, plan, step_size=20,
                               rng=np.random.default_rng(0))
    m0 = data.metric_names[0]
    assert out[m0].shape[-1] == 3                    # quantile triple
    assert np.isfinite(out[m0]).all() and (out[m0] > 0).all()
    # quantiles are monotone after serving-side calibration sort
    assert (np.diff(out[m0], axis=-1) >= 0).all()

    # sanity check: inject traffic-unjustified CPU burn, score residuals
    comp = app.all_components[1]
    series = np.asarray(data.resources[f"{comp}_cpu"], dtype=np.float64)
    q50 = series.copy()
    band = 0.2 * np.abs(series) + 1.0
    attacked = series.copy()
    attacked[30:50] += 10.0 * band[30:50]
    scorer = AnomalyScorer(threshold=0.25, min_run=3)
    report = scorer.score(attacked, q50 - band, q50, q50 + band)
    assert report.is_anomalous
    clean = scorer.score(series, q50 - band, q50, q50 + band)
    assert not clean.is_anomalous
