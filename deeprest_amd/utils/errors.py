"""Estimation-error reporting.

The reference's validation harness prints, per metric and per estimator, the
Median / 95th / 99th / Max of absolute errors on held-out windows
(reference: resource-estimation/estimate.py:112-122).  That comparison table
is the framework's accuracy contract; we reproduce it as structured data.
"""

from __future__ import annotations

from typing import Dict, Sequence

import numpy as np


def error_percentiles(abs_errors: Sequence[float]) -> Dict[str, float]:
    e = np.asarray(abs_errors, dtype=np.float64)
    if e.size == 0:
        return {"median": float("nan"), "p95": float("nan"),
                "p99": float("nan"), "max": float("nan")}
    return {
        "median": float(np.median(e)),
        "p95": float(np.percentile(e, 95)),
        "p99": float(np.percentile(e, 99)),
        "max": float(np.max(e)),
    }


def format_error_table(name: str, per_estimator: Dict[str, Dict[str, float]]) -> str:
    """Render one metric's estimator comparison in the reference's layout."""
    lines = [f"===== {name} ====="]
    for est, stats in per_estimator.items():
        lines.append(
            "   %-5s => Median: %.4f | 95-th: %.4f | 99-th: %.4f | Max: %.4f"
            % (est.upper(), stats["median"], stats["p95"], stats["p99"], stats["max"])
        )
    return "\n".join(lines)


def quantile_coverage(labels: np.ndarray, q_lo: np.ndarray,
                      q_hi: np.ndarray) -> Dict[str, float]:
    """Empirical calibration of a quantile band: the fraction of held-out
    observations inside [q_lo, q_hi] (nominal coverage for the (.05, .95)
    band is 0.90), plus the one-sided miss rates.  The reference never
    evaluates its bands; for the sanity-check use case (anomaly = utilization
    outside the band) calibration IS the operating characteristic."""
    y = np.asarray(labels, dtype=np.float64).ravel()
    lo = np.asarray(q_lo, dtype=np.float64).ravel()
    hi = np.asarray(q_hi, dtype=np.float64).ravel()
    if y.size == 0:
        return {"coverage": float("nan"), "below": float("nan"),
                "above": float("nan")}
    return {
        "coverage": float(np.mean((y >= lo) & (y <= hi))),
        "below": float(np.mean(y < lo)),
        "above": float(np.mean(y > hi)),
    }
