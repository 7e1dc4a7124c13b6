"""Model families exercised by the reference examples: GraphSAGE
(supervised + unsupervised link-pred), GAT, GCN."""
from typing import Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from .layers import GATConv, GCNConv, SAGEConv


def _layer_trim(num_sampled_nodes, num_sampled_edges, num_layers, layer):
    """Rows/edges needed by layer `layer` (0-based) of an L-layer GNN over a
    glt_amd multi-hop batch: layer l only needs nodes of hops
    0..L-l and edges of hops 1..L-l (hop-ordered concatenation).
    Returns (n_in_rows, n_edges, n_out_rows) or None when trim info is
    unusable."""
    if not num_sampled_nodes or not num_sampled_edges:
        return None
    nsn = [int(v) for v in num_sampled_nodes]
    nse = [int(v) for v in num_sampled_edges]
    L = num_layers
    if len(nse) < L or len(nsn) < L + 1:
        return None
    keep_hops = L - layer          # edges of hops 1..keep_hops
    n_edges = sum(nse[:keep_hops])
    n_in = sum(nsn[:keep_hops + 1])
    n_out = sum(nsn[:keep_hops])
    return n_in, n_edges, n_out


class GraphSAGE(nn.Module):
    def __init__(self, in_channels: int, hidden_channels: int,
                 num_layers: int, out_channels: Optional[int] = None,
                 dropout: float = 0.0):
        super().__init__()
        out_channels = out_channels or hidden_channels
        self.convs = nn.ModuleList()
        dims = ([in_channels] + [hidden_channels] * (num_layers - 1) +
                [out_channels])
        for i in range(num_layers):
            self.convs.append(SAGEConv(dims[i], dims[i + 1]))
        self.dropout = dropout

    def forward(self, x, edge_index, num_sampled_nodes=None,
                num_sampled_edges=None):
        """Hop-wise trimmed forward: with the per-hop counts from a sampled
        batch, layer l runs only over the rows/edges still reachable from
        the seeds — ~(fan-out) x less compute on the deep layers."""
        L = len(self.convs)
        for i, conv in enumerate(self.convs):
            act = i < L - 1  # fused into the projection GEMM epilogue
            trim = _layer_trim(num_sampled_nodes, num_sampled_edges, L, i)
            if trim is not None:
                n_in, n_edges, n_out = trim
                x = conv(x[:n_in], edge_index[:, :n_edges],
                         num_target=n_out, fuse_relu=act)
            else:
                x = conv(x, edge_index, fuse_relu=act)
            if act:
                x = F.dropout(x, p=self.dropout, training=self.training)
        return x


class GAT(nn.Module):
    def __init__(self, in_channels: int, hidden_channels: int,
                 num_layers: int, out_channels: Optional[int] = None,
                 heads: int = 4, dropout: float = 0.0):
        super().__init__()
        out_channels = out_channels or hidden_channels
        self.convs = nn.ModuleList()
        dim = in_channels
        for i in range(num_layers - 1):
            self.convs.append(GATConv(dim, hidden_channels, heads=heads))
            dim = hidden_channels * heads
        self.convs.append(GATConv(dim, out_channels, heads=1, concat=False))
        self.dropout = dropout

    def forward(self, x, edge_index, num_sampled_nodes=None,
                num_sampled_edges=None):
        L = len(self.convs)
        for i, conv in enumerate(self.convs):
            trim = _layer_trim(num_sampled_nodes, num_sampled_edges, L, i)
            if trim is not None:
                n_in, n_edges, n_out = trim
                x = conv(x[:n_in], edge_index[:, :n_edges],
                         num_target=n_out)
            else:
                x = conv(x, edge_index)
            if i < L - 1:
                x = F.elu(x)
                x = F.dropout(x, p=self.dropout, training=self.training)
        return x


class GCN(nn.Module):
    def __init__(self, in_channels: int, hidden_channels: int,
                 num_layers: int, out_channels: Optional[int] = None,
                 dropout: float = 0.0):
        super().__init__()
        out_channels = out_channels or hidden_channels
        self.convs = nn.ModuleList()
        dims = ([in_channels] + [hidden_channels] * (num_layers - 1) +
                [out_channels])
        for i in range(num_layers):
            self.convs.append(GCNConv(dims[i], dims[i + 1]))
        self.dropout = dropout

    def forward(self, x, edge_index, num_sampled_nodes=None,
                num_sampled_edges=None):
        L = len(self.convs)
        for i, conv in enumerate(self.convs):
            trim = _layer_trim(num_sampled_nodes, num_sampled_edges, L, i)
            if trim is not None:
                n_in, n_edges, n_out = trim
                x = conv(x[:n_in], edge_index[:, :n_edges],
                         num_target=n_out)
            else:
                x = conv(x, edge_index)
            if i < L - 1:
                x = F.relu(x)
                x = F.dropout(x, p=self.dropout, training=self.training)
        return x


def unsupervised_link_pred_loss(h: torch.Tensor,
                                edge_label_index: torch.Tensor,
                                edge_label: torch.Tensor) -> torch.Tensor:
    """Binary link-prediction loss on embedding dot products (the
    unsupervised GraphSAGE objective)."""
    src = h[edge_label_index[0]]
    dst = h[edge_label_index[1]]
    logits = (src * dst).sum(-1).float()  # bce in fp32 for bf16 runs
    return F.binary_cross_entropy_with_logits(logits, edge_label.float())
