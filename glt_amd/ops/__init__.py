from .segment import segment_mean, segment_mean_cat
from .linear import mfma_linear, use_mfma_linear
from .gat import gat_softmax_aggregate

__all__ = ["segment_mean", "segment_mean_cat", "mfma_linear", "use_mfma_linear", "gat_softmax_aggregate"]
