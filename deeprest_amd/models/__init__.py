from .baselines import ResourceAwareBaseline, ComponentAwareBaseline
from .net import DeepRestNet, DeepRestNetConfig

__all__ = [
    "ResourceAwareBaseline",
    "ComponentAwareBaseline",
    "DeepRestNet",
    "DeepRestNetConfig",
]
