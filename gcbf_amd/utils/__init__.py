from .amp import enable_bf16

__all__ = ["enable_bf16"]
