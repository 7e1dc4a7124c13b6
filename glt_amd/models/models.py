"""Model families exercised by the reference examples: GraphSAGE
(supervised + unsupervised link-pred), GAT, GCN."""
from typing import List, Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from .layers import GATConv, GCNConv, SAGEConv


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

    def forward(self, x, edge_index):
        for i, conv in enumerate(self.convs):
            x = conv(x, edge_index)
            if i < len(self.convs) - 1:
                x = F.relu(x)
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

    def forward(self, x, edge_index):
        for i, conv in enumerate(self.convs):
            x = conv(x, edge_index)
            if i < len(self.convs) - 1:
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

    def forward(self, x, edge_index):
        for i, conv in enumerate(self.convs):
            x = conv(x, edge_index)
            if i < len(self.convs) - 1:
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
    logits = (src * dst).sum(-1)
    return F.binary_cross_entropy_with_logits(logits, edge_label.float())
