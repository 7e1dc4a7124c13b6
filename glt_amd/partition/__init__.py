from .base import (FeaturePartitionData, GraphPartitionData,
                   PartitionerBase, cat_feature_cache, load_partition,
                   save_graph_cache, save_meta)
from .partition_book import (GLTPartitionBook, PartitionBook,
                             RangePartitionBook)
from .random_partitioner import RandomPartitioner
from .frequency_partitioner import FrequencyPartitioner

__all__ = [
    "FeaturePartitionData", "GraphPartitionData", "PartitionerBase",
    "cat_feature_cache", "load_partition", "save_graph_cache", "save_meta", "GLTPartitionBook",
    "PartitionBook", "RangePartitionBook", "RandomPartitioner",
    "FrequencyPartitioner",
]
