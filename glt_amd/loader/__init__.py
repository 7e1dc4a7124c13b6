from .transform import to_data, to_hetero_data
from .node_loader import NodeLoader
from .neighbor_loader import NeighborLoader
from .link_loader import LinkLoader, LinkNeighborLoader
from .subgraph_loader import SubGraphLoader

__all__ = ["to_data", "to_hetero_data", "NodeLoader", "NeighborLoader",
           "LinkLoader", "LinkNeighborLoader", "SubGraphLoader"]
