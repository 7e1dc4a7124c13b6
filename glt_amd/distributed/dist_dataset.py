"""DistDataset: one partition of a partitioned dataset (parity: reference
python/distributed/dist_dataset.py:85-317)."""
import os
from typing import Dict, List, Optional, Union

import torch

from ..data import Dataset, DeviceGroup, Feature, Graph, Topology
from ..partition import (PartitionBook, cat_feature_cache, load_partition)
from ..typing import EdgeType, NodeType


class DistDataset(Dataset):
    def __init__(self, num_partitions: int = 1, partition_idx: int = 0,
                 graph_partition=None, node_feature_partition=None,
                 edge_feature_partition=None, whole_node_labels=None,
                 node_pb=None, edge_pb=None, node_feat_pb=None,
                 edge_feat_pb=None, edge_dir: str = "out"):
        super().__init__(graph_partition, node_feature_partition,
                         edge_feature_partition, whole_node_labels,
                         edge_dir)
        self.num_partitions = num_partitions
        self.partition_idx = partition_idx
        self.node_pb = node_pb
        self.edge_pb = edge_pb
        # feature pbs can differ from graph pbs when a hot-cache was merged
        self._node_feat_pb = node_feat_pb
        self._edge_feat_pb = edge_feat_pb

    @property
    def node_feat_pb(self):
        return self._node_feat_pb if self._node_feat_pb is not None \
            else self.node_pb

    @property
    def edge_feat_pb(self):
        return self._edge_feat_pb if self._edge_feat_pb is not None \
            else self.edge_pb

    def load(self, root_dir: str, partition_idx: int,
             graph_mode: str = "ZERO_COPY",
             input_layout: str = "COO",
             feature_with_gpu: bool = True,
             graph_caching: bool = False,
             device_group_list: Optional[List[DeviceGroup]] = None,
             whole_node_label_file: Optional[Union[str, Dict]] = None,
             device: Optional[int] = None):
        """Load one partition saved in the GLT on-disk layout.

        graph_caching=True loads the whole-topology cache written by
        `partition.save_graph_cache` (every rank holds the full graph;
        only features stay partitioned — reference graph_caching mode).
        """
        (num_parts, graph_data, node_feat_data, edge_feat_data, node_pb,
         edge_pb) = load_partition(root_dir, partition_idx)
        if graph_caching:
            import os as _os

            from ..partition.base import _load_graph_dir

            whole = _load_graph_dir(_os.path.join(root_dir, "graph"))
            if whole is not None:
                graph_data = whole
        self.num_partitions = num_parts
        self.partition_idx = partition_idx
        self.node_pb = node_pb
        self.edge_pb = edge_pb

        target = "CSC" if self.edge_dir == "in" else "CSR"

        def build_graph(g):
            topo = Topology(g.edge_index, edge_ids=g.eids,
                            edge_weights=g.weights, input_layout="COO",
                            layout=target)
            return Graph(topo, graph_mode, device)

        if isinstance(graph_data, dict):
            self.graph = {et: build_graph(g) for et, g in graph_data.items()}
        elif graph_data is not None:
            self.graph = build_graph(graph_data)

        def build_feature(fp):
            if fp is None or fp.feats is None:
                return None, None
            feats, ids, id2index = cat_feature_cache(fp)
            return Feature(feats, split_ratio=1.0 if feature_with_gpu else 0,
                           device_group_list=device_group_list,
                           device=device, with_gpu=feature_with_gpu,
                           id2index=id2index), None

        if isinstance(node_feat_data, dict):
            self.node_features = {}
            for nt, fp in node_feat_data.items():
                f, _ = build_feature(fp)
                if f is not None:
                    self.node_features[nt] = f
        elif node_feat_data is not None:
            self.node_features, _ = build_feature(node_feat_data)

        if isinstance(edge_feat_data, dict):
            self.edge_features = {}
            for et, fp in edge_feat_data.items():
                f, _ = build_feature(fp)
                if f is not None:
                    self.edge_features[et] = f
        elif edge_feat_data is not None:
            self.edge_features, _ = build_feature(edge_feat_data)

        if whole_node_label_file is not None:
            if isinstance(whole_node_label_file, dict):
                self.node_labels = {
                    nt: torch.load(p, weights_only=False)
                    for nt, p in whole_node_label_file.items()}
            else:
                self.node_labels = torch.load(whole_node_label_file,
                                              weights_only=False)
        return self
