"""ODPS table-backed datasets (capability parity: reference
python/data/table_dataset.py:30-44, distributed/dist_table_dataset.py).

Streams (id, feature, label) records from ODPS tables through `common_io`
into an in-memory Dataset.  `common_io` is Alibaba-internal and not present
in this image; constructing these classes without it raises a clear error,
mirroring the reference's optional dependency behavior.
"""
from typing import List, Optional

import torch

from .dataset import Dataset

try:  # pragma: no cover - external dependency
    import common_io  # type: ignore

    _HAS_COMMON_IO = True
except ImportError:
    common_io = None
    _HAS_COMMON_IO = False


class TableDataset(Dataset):
    """Reads edge/node tables: edge table rows (src, dst[, weight]),
    node table rows (id, feature string '<v1>:<v2>:...'[, label])."""

    def __init__(self, edge_table: Optional[str] = None,
                 node_table: Optional[str] = None,
                 label_table: Optional[str] = None,
                 num_threads: int = 4, capacity: int = 1 << 16,
                 feature_delimiter: str = ":", **dataset_kwargs):
        if not _HAS_COMMON_IO:
            raise ImportError(
                "glt_amd.data.TableDataset requires the ODPS `common_io` "
                "package (Alibaba internal); install it or load data "
                "through glt_amd.data.Dataset builders instead")
        super().__init__(**dataset_kwargs)
        self.feature_delimiter = feature_delimiter
        if edge_table:
            self._load_edges(edge_table, num_threads, capacity)
        if node_table:
            self._load_nodes(node_table, num_threads, capacity)
        if label_table:
            self._load_labels(label_table, num_threads, capacity)

    def _read_all(self, table, capacity):
        reader = common_io.table.TableReader(table,
                                             capacity=capacity)
        records = []
        while True:
            try:
                records.extend(reader.read(capacity, allow_smaller=True))
            except common_io.exception.OutOfRangeException:
                break
        reader.close()
        return records

    def _load_edges(self, table, num_threads, capacity):
        recs = self._read_all(table, capacity)
        src = torch.tensor([int(r[0]) for r in recs])
        dst = torch.tensor([int(r[1]) for r in recs])
        self.init_graph(edge_index=torch.stack([src, dst]),
                        graph_mode="CPU")

    def _load_nodes(self, table, num_threads, capacity):
        recs = self._read_all(table, capacity)
        feats = torch.tensor(
            [[float(x) for x in r[1].split(self.feature_delimiter)]
             for r in recs])
        self.init_node_features(feats, with_gpu=False)

    def _load_labels(self, table, num_threads, capacity):
        recs = self._read_all(table, capacity)
        self.init_node_labels(torch.tensor([int(r[1]) for r in recs]))


class DistTableDataset(TableDataset):
    """Distributed ODPS table ingestion (capability parity: reference
    python/distributed/dist_table_dataset.py:149-353): every rank reads its
    slice of the tables, then the chunks are exchanged to their owning
    partitions with DistRandomPartitioner.  Requires `common_io`."""

    def __init__(self, edge_table=None, node_table=None, label_table=None,
                 num_nodes: int = 0, num_threads: int = 4,
                 capacity: int = 1 << 16, **kwargs):
        super().__init__(edge_table, node_table, label_table, num_threads,
                         capacity, **kwargs)
        from ..distributed.dist_random_partitioner import \
            DistRandomPartitioner

        rows, cols, _ = self.graph.topo.to_coo()
        feats = (self.node_features.cpu_tensor
                 if self.node_features is not None else None)
        self._partitioner = DistRandomPartitioner(
            num_nodes or self.graph.num_nodes,
            torch.stack([rows, cols]), local_node_feat=feats)

    def partition(self):
        return self._partitioner.partition()
