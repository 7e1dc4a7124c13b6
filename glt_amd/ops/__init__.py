from .segment import segment_mean
from .linear import mfma_linear, use_mfma_linear

__all__ = ["segment_mean", "mfma_linear", "use_mfma_linear"]
