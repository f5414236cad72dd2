"""REST ingestion + estimation API (north star: "REST ingestion API").

Endpoints (FastAPI; run with `uvicorn deeprest_amd.serve.api:create_app`):

- GET  /health                    liveness + native-extension status
- POST /ingest                    append raw-data windows (the L4 contract
                                  as JSON instead of pickle)
- GET  /ingest/stats              windows/apis/components ingested
- POST /featurize                 build the call-path feature space over the
                                  ingested windows
- POST /estimate                  what-if estimation for a traffic plan
                                  [{api: count}, ...] using the loaded model
- POST /predict                   raw (N, T, P) call-path count windows ->
                                  quantile predictions; with --micro-batch,
                                  concurrent requests coalesce into one
                                  hipGraph replay (serve/batcher.py)
- POST /anomaly                   sanity-check measured series against the
                                  model's quantile band
- GET  /apis                      known API endpoints (for what-if queries)
- GET  /results[/{exp}[/{comp}/{metric}]]  browse a results.pkl store (the
                                  reference web-demo's DataLoader surface,
                                  web-demo/dataloader.py:30-167, as REST)

The app holds an IngestStore plus an optional Predictor loaded from a
checkpoint; everything is JSON-serializable.
"""

from __future__ import annotations

from typing import Any, Dict, List, Optional

import numpy as np

from ..data.contract import ContractError, validate_raw_data
from ..data.featurize import Featurizer
from ..data.synthesizer import TraceSynthesizer
from .anomaly import AnomalyScorer
from .predictor import Predictor


class IngestStore:
    """In-memory raw-data window store behind the REST surface."""

    def __init__(self) -> None:
        self.windows: List[Dict[str, Any]] = []
        self.featurizer: Optional[Featurizer] = None
        self.synthesizer: Optional[TraceSynthesizer] = None

    def ingest(self, windows: List[Dict[str, Any]]) -> int:
        validate_raw_data(windows)
        self.windows.extend(windows)
        return len(self.windows)

    def featurize(self, model_space=None):
        """``model_space``: when a checkpointed model is loaded, its FROZEN
        call-path feature space — synthesized what-if vectors must line up
        with the model's input features, not with whatever subset the
        ingested windows happen to cover (unseen paths are dropped, exactly
        the reference's frozen-space extract semantics, featurize.py:27-33)."""
        if model_space is not None:
            self.featurizer = Featurizer(feature_space=model_space,
                                         use_native=False)
            data = self.featurizer.transform(self.windows)
            self.synthesizer = TraceSynthesizer(
                feature_space=model_space).fit(self.windows)
            return data
        self.featurizer = Featurizer(use_native=False).fit(self.windows)
        data = self.featurizer.transform(self.windows)
        self.synthesizer = TraceSynthesizer(
            feature_space=self.featurizer.feature_space
        ).fit(self.windows)
        return data


def create_app(checkpoint_path: Optional[str] = None, predictor: Optional[Predictor] = None,
               results_path: Optional[str] = None, micro_batch: bool = False,
               micro_batch_wait_ms: float = 2.0):
    from fastapi import FastAPI, HTTPException

    app = FastAPI(title="deeprest-amd", version="0.1.0")
    store = IngestStore()
    state = {"predictor": predictor, "results": None, "batcher": None}
    if checkpoint_path and predictor is None:
        state["predictor"] = Predictor.from_checkpoint(checkpoint_path)
    if micro_batch and state["predictor"] is not None:
        from .batcher import MicroBatcher

        state["batcher"] = MicroBatcher(
            state["predictor"],
            max_batch=max(state["predictor"].graph_batches),
            max_wait_ms=micro_batch_wait_ms)
    if results_path:
        from .results import ResultsStore

        state["results"] = ResultsStore.load(results_path)

    # ---- results browsing (the reference web-demo's DataLoader surface over
    # results.pkl — web-demo/dataloader.py:30-167 — as REST) ----
    def _results():
        rs = state["results"]
        if rs is None:
            raise HTTPException(status_code=400, detail="no results loaded "
                                "(serve with results_path=...)")
        return rs

    @app.get("/results")
    def results_index():
        return {"experiments": _results().experiments()}

    @app.get("/results/{experiment}")
    def results_experiment(experiment: str):
        rs = _results()
        if experiment not in rs.results:
            raise HTTPException(status_code=404, detail="unknown experiment")
        return {
            comp: sorted(metrics.keys())
            for comp, metrics in rs.results[experiment].items()
        }

    @app.get("/results/{experiment}/{component}/{metric}")
    def results_entry(experiment: str, component: str, metric: str):
        rs = _results()
        try:
            entry = rs.get(experiment, component, metric)
        except KeyError:
            raise HTTPException(status_code=404, detail="unknown entry")
        return {k: (v.tolist() if isinstance(v, np.ndarray) else v)
                for k, v in entry.items()}

    @app.get("/demo")
    def demo():
        from fastapi.responses import HTMLResponse

        from .demo import DEMO_HTML

        return HTMLResponse(DEMO_HTML)

    @app.get("/demo/results")
    def demo_results():
        """The reference web-demo's results browser (web-demo/app.py) over
        the /results REST surface."""
        from fastapi.responses import HTMLResponse

        from .demo import RESULTS_HTML

        return HTMLResponse(RESULTS_HTML)

    @app.get("/health")
    def health():
        from ..ops import native_available

        pred = state["predictor"]
        return {
            "status": "ok",
            "native_extension": native_available(),
            "model_loaded": pred is not None,
            # estimator-head observability: what transforms this server
            # applies on top of the net's quantile outputs
            "residual_base": bool(getattr(pred, "residual_ridge", None)
                                  is not None) if pred else None,
            "conformal_bands": bool(getattr(pred, "conformal", None)
                                    is not None) if pred else None,
            "target_transform": getattr(pred, "target_transform", None)
            if pred else None,
        }

    @app.post("/ingest")
    def ingest(payload: List[Dict[str, Any]]):
        try:
            total = store.ingest(payload)
        except ContractError as e:
            raise HTTPException(status_code=422, detail=str(e))
        return {"windows_total": total}

    @app.get("/ingest/stats")
    def ingest_stats():
        n = len(store.windows)
        apis = set()
        comps = set()
        for w in store.windows:
            for tr in w.get("traces", []):
                apis.add(f"{tr['component']}_{tr['operation']}")
            for m in w.get("metrics", []):
                comps.add(m["component"])
        return {"windows": n, "apis": sorted(apis), "components": sorted(comps)}

    @app.post("/featurize")
    def featurize():
        if not store.windows:
            raise HTTPException(status_code=400, detail="no ingested windows")
        pred = state["predictor"]
        model_space = pred.feature_space if pred is not None else None
        data = store.featurize(model_space=model_space)
        return {
            "num_paths": data.num_paths,
            "num_windows": data.num_windows,
            "metrics": data.metric_names,
            "frozen_to_model_space": model_space is not None,
        }

    @app.get("/apis")
    def apis():
        if store.synthesizer is None:
            raise HTTPException(status_code=400, detail="featurize first")
        return {"apis": store.synthesizer.apis}

    @app.post("/estimate")
    def estimate(payload: Dict[str, Any]):
        """payload: {'traffic_plan': [{api: count}, ...], 'seed': int?}"""
        pred: Optional[Predictor] = state["predictor"]
        if pred is None:
            raise HTTPException(status_code=400, detail="no model loaded")
        if store.synthesizer is None:
            raise HTTPException(status_code=400, detail="featurize first")
        plan = payload.get("traffic_plan")
        if not plan:
            raise HTTPException(status_code=422, detail="traffic_plan required")
        n_model = pred.model.spec.num_paths
        n_syn = len(store.synthesizer.feature_space)
        if n_syn != n_model:
            raise HTTPException(
                status_code=422,
                detail=f"feature space mismatch: synthesizer has {n_syn} call "
                       f"paths, model expects {n_model} — load the predictor "
                       "with its checkpoint feature space (POST /featurize "
                       "again after loading)")
        rng = np.random.default_rng(payload.get("seed"))
        try:
            out = pred.predict_what_if(
                store.synthesizer, plan,
                step_size=int(payload.get("step_size", 60)), rng=rng,
            )
        except KeyError as e:
            raise HTTPException(status_code=422, detail=str(e))
        return {
            "quantiles": [0.05, 0.50, 0.95],
            "predictions": {k: v.tolist() for k, v in out.items()},
        }

    @app.post("/predict")
    def predict_windows(payload: Dict[str, Any]):
        """payload: {'windows': (N, T, P) raw call-path count windows} ->
        per-metric quantile predictions. With micro-batching enabled,
        concurrent requests coalesce into one hipGraph replay."""
        pred: Optional[Predictor] = state["predictor"]
        if pred is None:
            raise HTTPException(status_code=400, detail="no model loaded")
        w = payload.get("windows")
        if w is None:
            raise HTTPException(status_code=422, detail="windows required")
        w = np.asarray(w, dtype=np.float64)
        if w.ndim != 3 or w.shape[-1] != pred.model.spec.num_paths:
            raise HTTPException(
                status_code=422,
                detail=f"windows must be (N, T, {pred.model.spec.num_paths})")
        target = state["batcher"] or pred
        out = target.predict(w)
        resp = {k: v.tolist() for k, v in out.items()}
        stats = None
        if state["batcher"] is not None:
            stats = {"batches_run": state["batcher"].batches_run,
                     "requests_served": state["batcher"].requests_served}
        return {"predictions": resp, "quantiles": list(pred.model.cfg.quantiles),
                "micro_batch": stats}

    @app.post("/anomaly")
    def anomaly(payload: Dict[str, Any]):
        """payload: {'measured': {metric: [..]}, 'predicted': {metric: [[q05,q50,q95],..]}}"""
        scorer = AnomalyScorer(
            threshold=float(payload.get("threshold", 0.25)),
            min_run=int(payload.get("min_run", 3)),
        )
        measured = {k: np.asarray(v) for k, v in payload["measured"].items()}
        preds = {k: np.asarray(v) for k, v in payload["predicted"].items()}
        reports = scorer.score_all(measured, preds)
        return {
            k: {
                "anomalous": r.is_anomalous,
                "windows": r.windows,
                "max_score": float(r.scores.max()) if len(r.scores) else 0.0,
            }
            for k, r in reports.items()
        }

    return app
