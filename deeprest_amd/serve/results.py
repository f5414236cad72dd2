"""Results store in the web-demo schema.

The durable artifact of an experiment run is results.pkl, whose schema is
defined by its reader (reference: web-demo/dataloader.py:56-140):

    results[experiment][component][metric] = {
        'calls': [per-API call-count series, ...],
        'measurement': [...],                    # ground-truth utilization
        'prediction_bl-resrc': [...],            # ResourceAware baseline
        'prediction_bl-api':   [...],            # ComponentAware baseline
        'prediction_bl-trace': [...],            # TraceAware baseline
        'prediction_ours':     [...],            # the estimation engine
        'scale_groundtruth': [per-query-window scale, ...],
        'scale_bl-resrc': [...], 'scale_bl-api': [...],
        'scale_bl-trace': [...], 'scale_ours': [...],
    }

We keep that exact key layout so the reference's demo could read our output.
Scale factors follow dataloader.py:112-118 semantics: predicted (or
measured) peak of each query window divided by the learning-period peak.
"""

from __future__ import annotations

import pickle
from typing import Dict, List, Optional, Sequence

import numpy as np

ESTIMATOR_KEYS = ("bl-resrc", "bl-api", "bl-trace", "ours")


def _scales(pred_windows: np.ndarray, train_peak: float,
            anchor: Optional[float] = None) -> List[float]:
    """Per query window: peak prediction / learning-period peak.

    ``anchor`` (monotone metrics: memory, disk usage) re-anchors each window
    to the last learning-period value before taking the peak — the
    reference's semantics (reference: web-demo/dataloader.py:143-156:
    ``pred - pred[0] + gt_offset``): for metrics that only ever grow, the
    absolute level is history, and only growth relative to where the
    learning period ended is a meaningful scale comparison."""
    train_peak = max(float(train_peak), 1e-9)
    out = []
    for w in pred_windows:
        if len(w) == 0:
            out.append(0.0)
            continue
        w = np.asarray(w, dtype=np.float64)
        if anchor is not None:
            w = w - w[0] + anchor
        out.append(float(np.max(w)) / train_peak)
    return out


def build_results_entry(
    measurement: np.ndarray,                     # (T_total,) ground truth series
    predictions: Dict[str, np.ndarray],          # est -> (K, W) query-window preds
    calls: Optional[Sequence[np.ndarray]] = None,  # per-API call series
    train_len: Optional[int] = None,             # learning-period length
    train_peak: Optional[float] = None,          # scale anchor override: for
    # scenario entries whose measurement is a SEPARATE query timeline, pass
    # the base learning period's peak (the reference's scale semantics) and
    # train_len=0 so predictions align from the series start
    reanchor: bool = False,                      # monotone metric (memory/
    # usage): re-anchor windows to the learning period's last value before
    # computing scale factors (dataloader.py:143-156)
    anchor_value: Optional[float] = None,        # the learning period's last
    # value when the measurement series here is a separate query timeline
) -> Dict[str, object]:
    measurement = np.asarray(measurement, dtype=np.float64)
    t_train = train_len if train_len is not None else len(measurement) // 2
    if train_peak is None:
        train_peak = float(np.max(measurement[:t_train])) if t_train > 0 else 1.0
    anchor = None
    if reanchor:
        anchor = anchor_value if anchor_value is not None else (
            float(measurement[t_train - 1]) if t_train > 0 else
            float(measurement[0]))

    entry: Dict[str, object] = {
        "calls": [list(np.asarray(c, dtype=np.float64)) for c in (calls or [])],
        "measurement": list(measurement),
    }
    K = W = None
    for est in ESTIMATOR_KEYS:
        if est not in predictions:
            continue
        pw = np.asarray(predictions[est], dtype=np.float64)
        if K is None:
            K, W = pw.shape
        entry[f"prediction_{est}"] = list(pw.reshape(-1))
        entry[f"scale_{est}"] = _scales(pw, train_peak, anchor)
    if K is not None:
        gt_scales = []
        for ki in range(K):
            seg = measurement[t_train + ki * W : t_train + (ki + 1) * W]
            if len(seg) == 0:
                gt_scales.append(0.0)
                continue
            if anchor is not None:
                seg = seg - seg[0] + anchor
            gt_scales.append(float(np.max(seg)) / max(train_peak, 1e-9))
        entry["scale_groundtruth"] = gt_scales
    return entry


class ResultsStore:
    """Nested experiment -> component -> metric -> entry store, picklable in
    the reference's on-disk format."""

    def __init__(self) -> None:
        self.results: Dict[str, Dict[str, Dict[str, dict]]] = {}

    def add(self, experiment: str, component: str, metric: str, entry: dict) -> None:
        self.results.setdefault(experiment, {}).setdefault(component, {})[metric] = entry

    def experiments(self) -> List[str]:
        return list(self.results.keys())

    def get(self, experiment: str, component: str, metric: str) -> dict:
        return self.results[experiment][component][metric]

    def save(self, path: str) -> None:
        with open(path, "wb") as f:
            pickle.dump(self.results, f)

    @staticmethod
    def load(path: str) -> "ResultsStore":
        store = ResultsStore()
        with open(path, "rb") as f:
            store.results = pickle.load(f)
        return store
