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


def _numpy_safe_globals() -> list:
    """Safe-unpickle allowlist for the numpy values our format carries
    (adjacency matrix, scaler arrays, numpy RNG state)."""
    globs: list = [np.ndarray, np.dtype]
    try:  # numpy >= 2
        from numpy._core import multiarray as _ma
    except ImportError:  # numpy 1.x
        from numpy.core import multiarray as _ma  # type: ignore
    globs.append(_ma._reconstruct)
    try:
        import numpy.dtypes as _npdt

        globs += [getattr(_npdt, n) for n in dir(_npdt) if n.endswith("DType")]
    except ImportError:
        pass
    return globs


def load_checkpoint(path: str, map_location="cpu",
                    trusted: bool = False) -> Dict[str, Any]:
    """Load a checkpoint with the weights-only unpickler by default.

    The format is pure tensors / containers / numpy arrays, so
    ``weights_only=True`` plus a numpy allowlist covers it — an untrusted
    file cannot run code through pickle.  Pass ``trusted=True`` only for
    checkpoints you produced yourself if one predates this format.
    """
    if trusted:
        return torch.load(path, map_location=map_location, weights_only=False)
    with torch.serialization.safe_globals(_numpy_safe_globals()):
        return torch.load(path, map_location=map_location, weights_only=True)
