"""Compute ops for the estimation engine.

Every hot op has two implementations:

- a hand-written CDNA4 HIP kernel (deeprest_amd/csrc/*.hip, built in-tree as
  ``deeprest_amd._C``) — the ONLY path used on a GPU;
- a plain PyTorch composition (``reference_*``) used on CPU for tests and as
  the numerics oracle the kernels are validated against.

On a CUDA (ROCm) tensor these ops refuse to fall back: if the native
extension is missing on a GPU machine they raise, so a silent eager fallback
can never masquerade as kernel coverage.
"""

from .native import native_available, require_native, load_native
from .gru import fused_gru_sequence, reference_gru_sequence
from .layernorm import layer_norm, reference_layer_norm
from .attention import mha_forward, reference_mha
from .pinball import pinball_loss, reference_pinball_loss
from .adam import fused_adam_step

__all__ = [
    "native_available",
    "require_native",
    "load_native",
    "fused_gru_sequence",
    "reference_gru_sequence",
    "layer_norm",
    "reference_layer_norm",
    "mha_forward",
    "reference_mha",
    "pinball_loss",
    "reference_pinball_loss",
    "fused_adam_step",
]
