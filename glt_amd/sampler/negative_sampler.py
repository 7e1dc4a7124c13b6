"""Random negative sampler (parity: reference python/sampler/negative_sampler.py)."""
from typing import Optional, Union

import torch

from ..data import Graph


class RandomNegativeSampler:
    """Samples (row, col) pairs absent from the graph.

    Args:
      graph: the Graph (CSR, edge_dir-aware: for 'in' graphs results are
        returned flipped back to (src, dst) order).
      mode: 'binary' semantics — strict rejection with `trials` attempts,
        optionally padded to exactly req_num with unchecked pairs.
    """

    def __init__(self, graph: Graph, trials: int = 5, padding: bool = False,
                 edge_dir: str = "out"):
        from .. import _C

        self._C = _C
        self.graph = graph
        self.trials = trials
        self.padding = padding
        self.edge_dir = edge_dir

    def sample(self, req_num: int) -> torch.Tensor:
        g = self.graph
        out = self._C.sample_negative(g.indptr, g.indices, g.num_nodes,
                                      req_num, self.trials, self.padding)
        if self.edge_dir == "in":
            out = out.flip(0)
        return out
