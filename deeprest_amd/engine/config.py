"""Typed configuration + CLI.

The reference configures everything through module-level constants edited in
place (reference: resource-estimation/estimate.py:13-18, featurize.py:6-7,
SURVEY.md section 5.6).  We replace that with one typed config (YAML-loadable,
CLI-overridable) preserving the same tunables.
"""

from __future__ import annotations

import dataclasses
from dataclasses import dataclass, field
from typing import Any, Dict, Optional

import yaml

from ..models.net import DeepRestNetConfig


@dataclass
class DataConfig:
    input_path: Optional[str] = None      # input.pkl (featurized) path
    raw_path: Optional[str] = None        # raw_data.pkl path (featurize first)
    step_size: int = 60                   # window length (estimate.py:18)
    split: float = 0.40                   # train fraction (estimate.py:17)
    target_transform: str = "none"        # "log1p": quantile-preserving
                                          # compression for unseen-scale
                                          # extrapolation (dataset.py)
    # synthetic app fallback (when no input/raw path given)
    synth_apis: int = 8
    synth_components: int = 12
    synth_windows_per_day: int = 240
    synth_days: int = 4
    synth_seed: int = 1234


@dataclass
class TrainConfig:
    epochs: int = 50                      # estimate.py:14
    batch_size: int = 32                  # estimate.py:15
    lr: float = 1e-3                      # estimate.py:16
    eval_cycles: int = 9                  # estimate.py:86-88
    baseline_epochs: int = 100            # baselines.py:57
    run_baselines: bool = True
    dtype: str = "bf16"                   # compute dtype on GPU
    graph_step: bool = False              # hipGraph-capture the train step
    seed: int = 0
    checkpoint_path: Optional[str] = None
    resume: bool = False
    log_every: int = 1
    eval_every: int = 1                   # per-epoch eval cadence (rank 0);
                                          # the final epoch always evaluates
    lr_schedule: str = "none"             # "none" (reference: fixed lr) or
                                          # "cosine" (decay to 5% over the run)
    conformal: float = 0.0                # >0: target coverage for split-
                                          # conformal band widening (CQR) on
                                          # held-out calibration windows,
                                          # e.g. 0.9 for the (.05,.95) band
    residual_base: str = "none"           # "trace-ridge": the net learns the
                                          # RESIDUAL over a closed-form ridge
                                          # on call-path features — the ridge
                                          # extrapolates unseen traffic scale
                                          # linearly, the net corrects
                                          # in-range nonlinearity (quantiles
                                          # are shift-equivariant). Use when
                                          # train rows >> call paths
                                          # (measured: DEEPR 3.9 vs 7.8 at
                                          # P=454; HURTS at P~rows, see
                                          # profiles/r02_unseen_traffic.md)


@dataclass
class EngineConfig:
    data: DataConfig = field(default_factory=DataConfig)
    train: TrainConfig = field(default_factory=TrainConfig)
    model: DeepRestNetConfig = field(default_factory=DeepRestNetConfig)

    def to_dict(self) -> Dict[str, Any]:
        return {
            "data": dataclasses.asdict(self.data),
            "train": dataclasses.asdict(self.train),
            "model": self.model.to_dict(),
        }

    @staticmethod
    def from_dict(d: Dict[str, Any]) -> "EngineConfig":
        cfg = EngineConfig()
        for section, cls in (("data", DataConfig), ("train", TrainConfig)):
            if section in d:
                known = {f.name for f in dataclasses.fields(cls)}
                setattr(cfg, section, cls(**{k: v for k, v in d[section].items() if k in known}))
        if "model" in d:
            m = dict(d["model"])
            if "quantiles" in m:
                m["quantiles"] = tuple(m["quantiles"])
            cfg.model = DeepRestNetConfig(**m)
        return cfg

    @staticmethod
    def load(path: str) -> "EngineConfig":
        with open(path) as f:
            return EngineConfig.from_dict(yaml.safe_load(f) or {})

    def save(self, path: str) -> None:
        with open(path, "w") as f:
            yaml.safe_dump(self.to_dict(), f)


def apply_cli_overrides(cfg: EngineConfig, overrides) -> EngineConfig:
    d = cfg.to_dict()
    for item in overrides:
        path, _, value = item.partition("=")
        section, _, key = path.partition(".")
        if section not in d or key not in d[section]:
            raise KeyError(f"unknown config key: {path}")
        current = d[section][key]
        d[section][key] = yaml.safe_load(value) if not isinstance(current, str) else value
    return EngineConfig.from_dict(d)
