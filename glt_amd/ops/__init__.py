from .segment import segment_mean

__all__ = ["segment_mean"]
