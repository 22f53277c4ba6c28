"""Extensions: privacy (DP + attacks), quantization, RL reweighting
(reference: extensions/)."""

from . import privacy  # noqa: F401
from .quantization import quant_arena, quant_model  # noqa: F401

__all__ = ["privacy", "quant_arena", "quant_model"]
