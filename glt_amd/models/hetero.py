"""Heterogeneous models: RGNN (rgat / rsage, as in the reference MLPerf
IGBH example, reference examples/igbh/rgnn.py capability) and a generic
HeteroConv combinator."""
from typing import Dict, List, Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..typing import EdgeType, NodeType
from .layers import GATConv, SAGEConv


class HeteroConv(nn.Module):
    """Applies a per-edge-type conv and sums contributions per dst type."""

    def __init__(self, convs: Dict[EdgeType, nn.Module], aggr: str = "sum"):
        super().__init__()
        self.convs = nn.ModuleDict({"__".join(k): v for k, v in convs.items()})
        self.aggr = aggr

    def forward(self, x_dict: Dict[NodeType, torch.Tensor],
                edge_index_dict: Dict[EdgeType, torch.Tensor]):
        out: Dict[NodeType, List[torch.Tensor]] = {}
        for etype, ei in edge_index_dict.items():
            key = "__".join(etype)
            if key not in self.convs:
                continue
            # target side of a glt_amd batch edge is etype-dependent: the
            # sampler emits edges keyed so that edge_index[0] is the
            # walked-from (seed-side) type.
            src_t, _, dst_t = etype
            tgt_t = src_t  # seed side
            x_tgt = x_dict.get(tgt_t)
            x_src = x_dict.get(dst_t)
            if x_tgt is None or x_src is None:
                continue
            conv = self.convs[key]
            try:
                h = conv((x_tgt, x_src), ei)
            except TypeError:
                h = conv_bipartite(conv, x_tgt, x_src, ei)
            out.setdefault(tgt_t, []).append(h)
        result = {}
        for t, hs in out.items():
            result[t] = torch.stack(hs).sum(0) if len(hs) > 1 else hs[0]
        return result


def conv_bipartite(conv, x_tgt, x_src, edge_index):
    """Run a homogeneous conv on a bipartite edge set by stacking target
    and source feature rows into one local space."""
    n_tgt = x_tgt.size(0)
    x = torch.cat([x_tgt, x_src], dim=0)
    ei = torch.stack([edge_index[0], edge_index[1] + n_tgt])
    return conv(x, ei)[:n_tgt]


class RGNN(nn.Module):
    """Relational GNN over hetero batches; model='rgat' or 'rsage'.

    Per layer: per-edge-type conv (GAT or SAGE) summed per target node
    type, plus a per-node-type self projection so every type advances to
    the layer's output dim even when it receives no messages; ReLU/dropout
    between layers; linear head on the predicted node type.
    """

    def __init__(self, etypes: List[EdgeType], in_dim: int, h_dim: int,
                 out_dim: int, num_layers: int = 2, n_heads: int = 4,
                 model: str = "rgat", dropout: float = 0.2,
                 node_types: Optional[List[NodeType]] = None):
        super().__init__()
        self.model = model
        if node_types is None:
            node_types = sorted({t for et in etypes for t in (et[0], et[2])})
        self.node_types = node_types
        self.layers = nn.ModuleList()
        self.self_lins = nn.ModuleList()
        dims = [in_dim] + [h_dim] * num_layers
        for li in range(num_layers):
            convs = {}
            for et in etypes:
                if model == "rgat":
                    convs[et] = GATConv(dims[li], dims[li + 1] // n_heads,
                                        heads=n_heads, concat=True)
                else:
                    convs[et] = SAGEConv(dims[li], dims[li + 1],
                                         root_weight=False)
            self.layers.append(HeteroConv(convs))
            self.self_lins.append(nn.ModuleDict(
                {t: nn.Linear(dims[li], dims[li + 1])
                 for t in node_types}))
        self.head = nn.Linear(h_dim, out_dim)
        self.dropout = dropout

    def forward(self, x_dict: Dict[NodeType, torch.Tensor],
                edge_index_dict: Dict[EdgeType, torch.Tensor],
                predict_type: Optional[NodeType] = None):
        h = x_dict
        for i, layer in enumerate(self.layers):
            h_conv = layer(h, edge_index_dict)
            h_next = {}
            for t, v in h.items():
                if t not in self.self_lins[i]:
                    continue
                out = self.self_lins[i][t](v)
                hc = h_conv.get(t)
                if hc is not None:
                    # conv output may cover fewer rows than the projection
                    # (message targets only); add on the common prefix
                    m = min(hc.size(0), out.size(0))
                    out = torch.cat([out[:m] + hc[:m], out[m:]], dim=0)
                h_next[t] = F.dropout(F.relu(out), p=self.dropout,
                                      training=self.training)
            h = h_next
        if predict_type is not None:
            return self.head(h[predict_type])
        return {t: self.head(v) for t, v in h.items()}
