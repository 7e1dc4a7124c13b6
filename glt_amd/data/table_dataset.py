"""ODPS table-backed datasets (parity: reference
python/data/table_dataset.py:30-44, distributed/dist_table_dataset.py:
149-353 threaded streaming readers).

Records stream through `num_threads` reader threads — each owns a table
slice and pushes bounded chunks into a queue while the consumer converts
chunks to tensors incrementally — so ingestion overlaps parsing and peak
memory is O(capacity), not O(table).

The reader backend is injectable (`reader_factory`): the default binds
ODPS `common_io` (Alibaba-internal, absent from this image — raises a
clear ImportError, mirroring the reference's optional dependency), and
tests inject in-memory readers so the whole streaming pipeline is
exercised in CI without ODPS.
"""
import queue
import threading
from typing import Callable, List, Optional

import torch

from .dataset import Dataset

try:  # pragma: no cover - external dependency
    import common_io  # type: ignore

    _HAS_COMMON_IO = True
except ImportError:
    common_io = None
    _HAS_COMMON_IO = False


def _common_io_factory(table: str, slice_id: int, slice_count: int,
                       capacity: int):  # pragma: no cover - needs ODPS
    return common_io.table.TableReader(table, slice_id=slice_id,
                                       slice_count=slice_count,
                                       capacity=capacity)


class _SliceReader(threading.Thread):
    """One table slice -> chunks into the shared bounded queue."""

    def __init__(self, factory, table, slice_id, slice_count, capacity,
                 out_q):
        super().__init__(daemon=True,
                         name=f"glt-table-reader-{slice_id}")
        self.factory = factory
        self.table = table
        self.slice_id = slice_id
        self.slice_count = slice_count
        self.capacity = capacity
        self.q = out_q
        self.error = None

    def run(self):
        reader = None
        try:
            reader = self.factory(self.table, self.slice_id,
                                  self.slice_count, self.capacity)
            while True:
                try:
                    recs = reader.read(self.capacity, allow_smaller=True)
                except StopIteration:
                    break
                except Exception as e:  # common_io OutOfRangeException
                    if type(e).__name__ == "OutOfRangeException":
                        break
                    raise
                if not recs:
                    break
                self.q.put(recs)
        except Exception as e:  # noqa: BLE001
            self.error = e
        finally:
            if reader is not None:
                try:
                    reader.close()
                except Exception:
                    pass
            self.q.put(None)  # slice end marker


def stream_table(table: str, num_threads: int, capacity: int,
                 reader_factory: Optional[Callable] = None):
    """Yield record chunks from `num_threads` parallel slice readers in
    arrival order.  Chunk = whatever the backend returns from one
    read(capacity) call (a list of record tuples)."""
    if reader_factory is None:
        if not _HAS_COMMON_IO:
            raise ImportError(
                "glt_amd.data table streaming requires the ODPS "
                "`common_io` package (Alibaba internal) or an explicit "
                "reader_factory; load data through glt_amd.data.Dataset "
                "builders instead")
        reader_factory = _common_io_factory
    num_threads = max(1, int(num_threads))
    q: "queue.Queue" = queue.Queue(maxsize=num_threads * 2)
    readers = [_SliceReader(reader_factory, table, i, num_threads,
                            capacity, q) for i in range(num_threads)]
    for r in readers:
        r.start()
    live = len(readers)
    while live:
        chunk = q.get()
        if chunk is None:
            live -= 1
            continue
        yield chunk
    for r in readers:
        r.join(timeout=10)
        if r.error is not None:
            raise r.error


class TableDataset(Dataset):
    """Reads edge/node tables: edge table rows (src, dst[, weight]),
    node table rows (id, feature string '<v1>:<v2>:...'[, label])."""

    def __init__(self, edge_table: Optional[str] = None,
                 node_table: Optional[str] = None,
                 label_table: Optional[str] = None,
                 num_threads: int = 4, capacity: int = 1 << 16,
                 feature_delimiter: str = ":",
                 reader_factory: Optional[Callable] = None,
                 **dataset_kwargs):
        if reader_factory is None and not _HAS_COMMON_IO:
            raise ImportError(
                "glt_amd.data.TableDataset requires the ODPS `common_io` "
                "package (Alibaba internal) or a reader_factory; load "
                "data through glt_amd.data.Dataset builders instead")
        super().__init__(**dataset_kwargs)
        self.feature_delimiter = feature_delimiter
        self._factory = reader_factory
        if edge_table:
            self._load_edges(edge_table, num_threads, capacity)
        if node_table:
            self._load_nodes(node_table, num_threads, capacity)
        if label_table:
            self._load_labels(label_table, num_threads, capacity)

    # chunks convert to tensors as they arrive; one cat at the end
    def _load_edges(self, table, num_threads, capacity):
        srcs: List[torch.Tensor] = []
        dsts: List[torch.Tensor] = []
        weights: List[torch.Tensor] = []
        has_w = False
        for recs in stream_table(table, num_threads, capacity,
                                 self._factory):
            srcs.append(torch.tensor([int(r[0]) for r in recs]))
            dsts.append(torch.tensor([int(r[1]) for r in recs]))
            if recs and len(recs[0]) > 2:
                has_w = True
                weights.append(torch.tensor([float(r[2]) for r in recs]))
        src = torch.cat(srcs) if srcs else torch.empty(0, dtype=torch.long)
        dst = torch.cat(dsts) if dsts else torch.empty(0, dtype=torch.long)
        self.init_graph(
            edge_index=torch.stack([src, dst]),
            edge_weights=torch.cat(weights) if has_w else None,
            graph_mode="CPU")

    def _load_nodes(self, table, num_threads, capacity):
        ids: List[torch.Tensor] = []
        feats: List[torch.Tensor] = []
        for recs in stream_table(table, num_threads, capacity,
                                 self._factory):
            ids.append(torch.tensor([int(r[0]) for r in recs]))
            feats.append(torch.tensor(
                [[float(x) for x in r[1].split(self.feature_delimiter)]
                 for r in recs]))
        if not ids:
            return
        all_ids = torch.cat(ids)
        all_feats = torch.cat(feats)
        # table slices arrive out of order: restore id order so row i
        # holds node i's features (ids are 0..N-1 in the reference format)
        order = torch.argsort(all_ids)
        self.init_node_features(all_feats[order], with_gpu=False)

    def _load_labels(self, table, num_threads, capacity):
        ids: List[torch.Tensor] = []
        labels: List[torch.Tensor] = []
        for recs in stream_table(table, num_threads, capacity,
                                 self._factory):
            ids.append(torch.tensor([int(r[0]) for r in recs]))
            labels.append(torch.tensor([int(r[1]) for r in recs]))
        if not ids:
            return
        all_ids = torch.cat(ids)
        all_labels = torch.cat(labels)
        self.init_node_labels(all_labels[torch.argsort(all_ids)])


class DistTableDataset(TableDataset):
    """Distributed ODPS table ingestion (parity: reference
    python/distributed/dist_table_dataset.py:149-353): every rank's
    slice readers stream its shard of the tables, then the chunks are
    exchanged to their owning partitions with DistRandomPartitioner."""

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
