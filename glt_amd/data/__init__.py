from .graph import Topology, Graph
from .feature import Feature, DeviceGroup
from .dataset import Dataset
from .reorder import sort_by_in_degree
from .unified_tensor import UnifiedTensor
from .table_dataset import TableDataset
from .xgmi_feature import XgmiShardedFeature
from . import vineyard_utils

__all__ = ["Topology", "Graph", "Feature", "DeviceGroup", "Dataset",
           "sort_by_in_degree", "UnifiedTensor", "TableDataset", "XgmiShardedFeature",
           "vineyard_utils"]
