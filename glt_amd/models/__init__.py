from .layers import GATConv, GCNConv, SAGEConv
from .models import GAT, GCN, GraphSAGE, unsupervised_link_pred_loss
from .hetero import HeteroConv, RGNN

__all__ = ["GATConv", "GCNConv", "SAGEConv", "GAT", "GCN", "GraphSAGE",
           "unsupervised_link_pred_loss", "HeteroConv", "RGNN"]
