import sys
import torch
sys.path.insert(0, ".")
import numpy as np
from deeprest_amd.data.synthetic import SyntheticApp, SyntheticAppConfig
from deeprest_amd.engine.config import DataConfig, EngineConfig, TrainConfig
from deeprest_amd.engine.trainer import Trainer
from deeprest_amd.models.net import DeepRestNetConfig

dev = torch.device("cuda")
app = SyntheticApp(SyntheticAppConfig(
    n_apis=13, n_components=12, windows_per_day=240, n_days=8,
    resource_noise=0.03, seed=77))
data = app.generate_featurized()
for residual in ("none", "trace-ridge"):
    for ep in (50, 100):
        cfg = EngineConfig(
            data=DataConfig(step_size=60, split=0.40),
            train=TrainConfig(epochs=ep, batch_size=32, lr=1e-3,
                              eval_cycles=9, baseline_epochs=100,
                              log_every=0, eval_every=5, graph_step=True,
                              residual_base=residual))
        cfg.model = DeepRestNetConfig(dropout=0.1)
        torch.manual_seed(0)
        tr = Trainer(data, cfg, device=dev)
        res = tr.train()
        med = {k: [] for k in ("resrc", "comp", "deepr")}
        wins = 0
        for per in res.error_tables.values():
            for k in med: med[k].append(per[k]["median"])
            wins += per["deepr"]["median"] <= per["comp"]["median"]
        print(residual, ep,
              {k: round(float(np.mean(v)), 3) for k, v in med.items()},
              "beats_comp:", wins, flush=True)
