"""Training visualizations (parity with the reference's matplotlib output).

The reference renders a learning curve and per-window overlays of ground
truth vs the three estimators at the end of training
(reference: resource-estimation/estimate.py:125-169, README.md:101-104).
Headless-safe (Agg backend); files instead of plt.show().
"""

from __future__ import annotations

from typing import Dict, Optional, Sequence

import numpy as np


def _plt():
    import matplotlib

    matplotlib.use("Agg")
    import matplotlib.pyplot as plt

    return plt


def plot_learning_curves(train_losses: Sequence[float],
                         test_losses: Sequence[float], path: str) -> str:
    plt = _plt()
    plt.figure(figsize=(6, 4))
    plt.title("Learning Curves")
    plt.plot(range(len(train_losses)), train_losses, label="Train", marker="x")
    plt.plot(range(len(test_losses)), test_losses, label="Test", marker="x")
    plt.xlabel("Epoch")
    plt.ylabel("Loss")
    plt.legend()
    plt.tight_layout()
    plt.savefig(path)
    plt.close()
    return path


def plot_estimator_overlay(
    metric_name: str,
    ground_truth: np.ndarray,
    predictions: Dict[str, np.ndarray],
    path: str,
    band: Optional[tuple] = None,
) -> str:
    """One window's ground truth vs every estimator; `band` optionally gives
    (q05, q95) for the engine's uncertainty band."""
    plt = _plt()
    plt.figure(figsize=(7, 4))
    plt.title(metric_name)
    plt.plot(ground_truth, label="Ground Truth", linestyle="--", color="red")
    colors = {"resrc": "green", "comp": "orange", "trace": "purple",
              "deepr": "mediumblue", "ours": "mediumblue"}
    for name, series in predictions.items():
        plt.plot(series, label=name, color=colors.get(name))
    if band is not None:
        q05, q95 = band
        plt.fill_between(range(len(q05)), q05, q95, alpha=0.2,
                         color="mediumblue", label="q05..q95")
    plt.legend()
    plt.tight_layout()
    plt.savefig(path)
    plt.close()
    return path
