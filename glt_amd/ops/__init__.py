from .segment import segment_mean, segment_mean_cat
from .linear import cast_linear, mfma_linear, use_mfma_linear
from .gat import gat_softmax_aggregate

__all__ = ["segment_mean", "segment_mean_cat", "cast_linear",
           "mfma_linear", "use_mfma_linear", "gat_softmax_aggregate"]
