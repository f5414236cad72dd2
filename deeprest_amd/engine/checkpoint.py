"""Checkpoint / resume.

The reference never persists training state (no
torch.save anywhere).  Our checkpoints carry everything inference and resume
need: model weights + spec, optimizer state, the fitted min-max scalers, the
call-path feature space M, epoch counter and RNG states.
"""

from __future__ import annotations

import os
from typing import Any, Dict, Optional

import numpy as np
import torch


def save_checkpoint(
    path: str,
    model,
    optimizer: Optional[torch.optim.Optimizer] = None,
    scaler_state: Optional[dict] = None,
    feature_space_state: Optional[dict] = None,
    epoch: int = 0,
    extra: Optional[Dict[str, Any]] = None,
) -> None:
    state = {
        "model": model.full_state(),
        "optimizer": optimizer.state_dict() if optimizer is not None else None,
        "scalers": scaler_state,
        "feature_space": feature_space_state,
        "epoch": epoch,
        "rng": {
            "torch": torch.get_rng_state(),
            "numpy": np.random.get_state(),
        },
        "extra": extra or {},
    }
    tmp = path + ".tmp"
    torch.save(state, tmp)
    os.replace(tmp, path)  # atomic: a crash mid-save never corrupts the last good checkpoint


def load_checkpoint(path: str, map_location="cpu") -> Dict[str, Any]:
    return torch.load(path, map_location=map_location, weights_only=False)
