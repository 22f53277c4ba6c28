from .quant import quant_arena, quant_model

__all__ = ["quant_arena", "quant_model"]
