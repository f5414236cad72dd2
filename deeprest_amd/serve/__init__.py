from .predictor import Predictor
from .results import ResultsStore, build_results_entry
from .anomaly import AnomalyScorer

__all__ = ["Predictor", "ResultsStore", "build_results_entry", "AnomalyScorer"]
