from .contract import Span, Window, validate_raw_data, load_raw_data, save_raw_data
from .featurize import FeatureSpace, Featurizer, FeaturizedData
from .synthesizer import TraceSynthesizer
from .synthetic import SyntheticApp, SyntheticAppConfig
from .windows import sliding_window, minmax_fit, minmax_apply, MinMaxScaler

__all__ = [
    "Span",
    "Window",
    "validate_raw_data",
    "load_raw_data",
    "save_raw_data",
    "FeatureSpace",
    "Featurizer",
    "FeaturizedData",
    "TraceSynthesizer",
    "SyntheticApp",
    "SyntheticAppConfig",
    "sliding_window",
    "minmax_fit",
    "minmax_apply",
    "MinMaxScaler",
]
