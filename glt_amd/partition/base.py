"""Offline graph partitioning: base partitioner, on-disk format, loaders.

Format parity with the reference so its partitioned datasets are drop-in
(reference python/partition/base.py:459-533 layout doc):

    root_dir/META                     (pickled dict: num_parts, data_cls, ...)
    root_dir/node_pb.pt | node_pb/<ntype>.pt
    root_dir/edge_pb.pt | edge_pb/<etype>.pt
    root_dir/part<i>/graph[/<etype>]/{rows,cols,eids[,weights]}.pt
    root_dir/part<i>/node_feat[/<ntype>]/{feats.pkl,ids.pkl
                                          [,cache_feats.pt,cache_ids.pt]}
    root_dir/part<i>/edge_feat[/<etype>]/{feats.pkl,ids.pkl,...}

feats.pkl/ids.pkl are sequences of pickled tensor chunks (appended), as in
reference utils/common.py:138-167.
"""
import os
import pickle
from dataclasses import dataclass
from typing import Dict, List, Optional, Tuple, Union

import torch

from ..typing import EdgeType, NodeType, as_str
from ..utils.common import ensure_dir
from .partition_book import GLTPartitionBook, PartitionBook


# ---------------------------------------------------------------------------
# chunked tensor files
# ---------------------------------------------------------------------------

def append_tensor_to_file(filename: str, tensor: torch.Tensor):
    with open(filename, "ab") as f:
        pickle.dump(tensor, f, pickle.HIGHEST_PROTOCOL)


def load_and_concatenate_tensors(filename: str, device=None):
    chunks = []
    with open(filename, "rb") as f:
        while True:
            try:
                chunks.append(pickle.load(f))
            except EOFError:
                break
    if not chunks:
        return None
    out = torch.empty((sum(c.shape[0] for c in chunks), *chunks[0].shape[1:]),
                      dtype=chunks[0].dtype, device=device)
    pos = 0
    for c in chunks:
        out[pos:pos + c.shape[0]] = c.to(device) if device else c
        pos += c.shape[0]
    return out


# ---------------------------------------------------------------------------
# partition data records
# ---------------------------------------------------------------------------

@dataclass
class GraphPartitionData:
    edge_index: torch.Tensor  # [2, E] (row, col) global ids
    eids: torch.Tensor
    weights: Optional[torch.Tensor] = None


@dataclass
class FeaturePartitionData:
    feats: torch.Tensor
    ids: torch.Tensor
    cache_feats: Optional[torch.Tensor] = None
    cache_ids: Optional[torch.Tensor] = None


# ---------------------------------------------------------------------------
# save helpers (same file layout as reference partition/base.py:43-170)
# ---------------------------------------------------------------------------

def save_meta(output_dir, num_parts, data_cls="homo", node_types=None,
              edge_types=None):
    meta = {"num_parts": num_parts, "data_cls": data_cls,
            "node_types": node_types, "edge_types": edge_types}
    with open(os.path.join(output_dir, "META"), "wb") as f:
        pickle.dump(meta, f, pickle.HIGHEST_PROTOCOL)


def save_node_pb(output_dir, node_pb, ntype=None):
    if ntype is not None:
        sub = os.path.join(output_dir, "node_pb")
        ensure_dir(sub)
        torch.save(node_pb, os.path.join(sub, f"{as_str(ntype)}.pt"))
    else:
        torch.save(node_pb, os.path.join(output_dir, "node_pb.pt"))


def save_edge_pb(output_dir, edge_pb, etype=None):
    if etype is not None:
        sub = os.path.join(output_dir, "edge_pb")
        ensure_dir(sub)
        torch.save(edge_pb, os.path.join(sub, f"{as_str(etype)}.pt"))
    else:
        torch.save(edge_pb, os.path.join(output_dir, "edge_pb.pt"))


def save_graph_partition(output_dir, partition_idx,
                         part: GraphPartitionData, etype=None):
    sub = os.path.join(output_dir, f"part{partition_idx}", "graph")
    if etype is not None:
        sub = os.path.join(sub, as_str(etype))
    ensure_dir(sub)
    torch.save(part.edge_index[0], os.path.join(sub, "rows.pt"))
    torch.save(part.edge_index[1], os.path.join(sub, "cols.pt"))
    torch.save(part.eids, os.path.join(sub, "eids.pt"))
    if part.weights is not None:
        torch.save(part.weights, os.path.join(sub, "weights.pt"))


def save_feature_partition(output_dir, partition_idx,
                           part: FeaturePartitionData, group="node_feat",
                           graph_type=None, chunk_rows: int = 1 << 20):
    sub = os.path.join(output_dir, f"part{partition_idx}", group)
    if graph_type is not None:
        sub = os.path.join(sub, as_str(graph_type))
    ensure_dir(sub)
    fpath = os.path.join(sub, "feats.pkl")
    ipath = os.path.join(sub, "ids.pkl")
    for p in (fpath, ipath):
        if os.path.exists(p):
            os.remove(p)
    for s in range(0, part.feats.shape[0], chunk_rows):
        append_tensor_to_file(fpath, part.feats[s:s + chunk_rows])
        append_tensor_to_file(ipath, part.ids[s:s + chunk_rows])
    if part.cache_feats is not None:
        torch.save(part.cache_feats, os.path.join(sub, "cache_feats.pt"))
        torch.save(part.cache_ids, os.path.join(sub, "cache_ids.pt"))


def save_graph_cache(output_dir, graph_partition_list, etype=None,
                     with_edge_feat: bool = False):
    """Whole-topology cache for `graph_caching=True` datasets (parity:
    reference partition/base.py:93-118): concatenates every partition's COO
    into one root-level graph/ directory."""
    if not graph_partition_list:
        return
    sub = os.path.join(output_dir, "graph")
    if etype is not None:
        sub = os.path.join(sub, as_str(etype))
    ensure_dir(sub)
    rows = torch.cat([g.edge_index[0] for g in graph_partition_list])
    cols = torch.cat([g.edge_index[1] for g in graph_partition_list])
    torch.save(rows, os.path.join(sub, "rows.pt"))
    torch.save(cols, os.path.join(sub, "cols.pt"))
    if with_edge_feat:
        torch.save(torch.cat([g.eids for g in graph_partition_list]),
                   os.path.join(sub, "eids.pt"))
    if graph_partition_list[0].weights is not None:
        torch.save(torch.cat([g.weights for g in graph_partition_list]),
                   os.path.join(sub, "weights.pt"))


# ---------------------------------------------------------------------------
# partitioner
# ---------------------------------------------------------------------------

class PartitionerBase:
    """Chunked offline partitioner (homo + hetero).

    Subclasses implement `_partition_node_ids(ntype) -> (ids_list, node_pb)`
    and may override `_cache_node_ids(ntype, partition_idx)` to attach a hot
    feature cache per partition.
    """

    def __init__(self, output_dir: str, num_parts: int,
                 num_nodes: Union[int, Dict[NodeType, int]],
                 edge_index: Union[torch.Tensor, Dict[EdgeType,
                                                      torch.Tensor]],
                 node_feat=None, edge_feat=None, edge_weights=None,
                 edge_assign_strategy: str = "by_src",
                 chunk_size: int = 10_000_000):
        self.output_dir = output_dir
        ensure_dir(output_dir)
        self.num_parts = num_parts
        self.num_nodes = num_nodes
        self.edge_index = edge_index
        self.node_feat = node_feat
        self.edge_feat = edge_feat
        self.edge_weights = edge_weights
        assert edge_assign_strategy in ("by_src", "by_dst")
        self.edge_assign_strategy = edge_assign_strategy
        self.chunk_size = chunk_size
        self.data_cls = "hetero" if isinstance(edge_index, dict) else "homo"
        if self.data_cls == "hetero":
            self.node_types = sorted(
                {t for et in edge_index for t in (et[0], et[2])})
            self.edge_types = list(edge_index.keys())
        else:
            self.node_types = None
            self.edge_types = None

    # -- abstract -----------------------------------------------------------
    def _partition_node_ids(self, ntype=None):
        raise NotImplementedError

    def _cache_node_ids(self, ntype, partition_idx):
        return None  # no cache by default

    # -- driver -------------------------------------------------------------
    def partition(self):
        if self.data_cls == "hetero":
            node_pbs = {}
            for nt in self.node_types:
                ids_list, node_pb = self._partition_node_ids(nt)
                save_node_pb(self.output_dir, node_pb, nt)
                node_pbs[nt] = node_pb
                nf = (self.node_feat or {}).get(nt)
                for p in range(self.num_parts):
                    self._save_node_feat_partition(nf, ids_list[p], nt, p)
            for et in self.edge_types:
                edge_pb = self._partition_edges(node_pbs, et)
                save_edge_pb(self.output_dir, edge_pb, et)
            save_meta(self.output_dir, self.num_parts, self.data_cls,
                      self.node_types, self.edge_types)
        else:
            ids_list, node_pb = self._partition_node_ids()
            save_node_pb(self.output_dir, node_pb)
            for p in range(self.num_parts):
                self._save_node_feat_partition(self.node_feat, ids_list[p],
                                               None, p)
            edge_pb = self._partition_edges({None: node_pb}, None)
            save_edge_pb(self.output_dir, edge_pb)
            save_meta(self.output_dir, self.num_parts, self.data_cls)

    def _partition_edges(self, node_pbs, etype) -> PartitionBook:
        ei = (self.edge_index[etype] if etype is not None
              else self.edge_index)
        ew = None
        if self.edge_weights is not None:
            ew = (self.edge_weights.get(etype)
                  if isinstance(self.edge_weights, dict)
                  else self.edge_weights)
        num_edges = ei.size(1)
        edge_pb = torch.zeros(num_edges, dtype=torch.uint8)
        if etype is not None:
            src_t, _, dst_t = etype
            assign_t = src_t if self.edge_assign_strategy == "by_src" \
                else dst_t
            pb = node_pbs[assign_t]
        else:
            pb = node_pbs[None]
        assign_nodes = ei[0] if self.edge_assign_strategy == "by_src" \
            else ei[1]
        ef = None
        if self.edge_feat is not None:
            ef = (self.edge_feat.get(etype)
                  if isinstance(self.edge_feat, dict) else self.edge_feat)
        for s in range(0, num_edges, self.chunk_size):
            e = min(s + self.chunk_size, num_edges)
            edge_pb[s:e] = pb[assign_nodes[s:e]].to(torch.uint8)
        for p in range(self.num_parts):
            mask = edge_pb == p
            eids = torch.nonzero(mask).flatten()
            part = GraphPartitionData(
                edge_index=ei[:, eids], eids=eids,
                weights=ew[eids] if ew is not None else None)
            save_graph_partition(self.output_dir, p, part, etype)
            if ef is not None:
                save_feature_partition(
                    self.output_dir, p,
                    FeaturePartitionData(feats=ef[eids], ids=eids),
                    group="edge_feat", graph_type=etype)
        return GLTPartitionBook(edge_pb)

    def _save_node_feat_partition(self, node_feat, ids, ntype, p):
        if node_feat is None:
            return
        cache_ids = self._cache_node_ids(ntype, p)
        part = FeaturePartitionData(
            feats=node_feat[ids], ids=ids,
            cache_feats=node_feat[cache_ids]
            if cache_ids is not None else None,
            cache_ids=cache_ids)
        save_feature_partition(self.output_dir, p, part, group="node_feat",
                               graph_type=ntype)


# ---------------------------------------------------------------------------
# loading
# ---------------------------------------------------------------------------

def _load_feature_dir(sub, device=None):
    if not os.path.isdir(sub):
        return None
    feats = load_and_concatenate_tensors(os.path.join(sub, "feats.pkl"))
    ids = load_and_concatenate_tensors(os.path.join(sub, "ids.pkl"))
    cache_feats = cache_ids = None
    if os.path.exists(os.path.join(sub, "cache_feats.pt")):
        cache_feats = torch.load(os.path.join(sub, "cache_feats.pt"),
                                 weights_only=False)
        cache_ids = torch.load(os.path.join(sub, "cache_ids.pt"),
                               weights_only=False)
    return FeaturePartitionData(feats=feats, ids=ids,
                                cache_feats=cache_feats,
                                cache_ids=cache_ids)


def _load_graph_dir(sub):
    if not os.path.isdir(sub):
        return None
    rows = torch.load(os.path.join(sub, "rows.pt"), weights_only=False)
    cols = torch.load(os.path.join(sub, "cols.pt"), weights_only=False)
    epath = os.path.join(sub, "eids.pt")
    # save_graph_cache writes eids only with_edge_feat=True; synthesize
    # positional ids otherwise
    eids = torch.load(epath, weights_only=False) \
        if os.path.exists(epath) else torch.arange(rows.numel())
    wpath = os.path.join(sub, "weights.pt")
    weights = torch.load(wpath, weights_only=False) \
        if os.path.exists(wpath) else None
    return GraphPartitionData(edge_index=torch.stack([rows, cols]),
                              eids=eids, weights=weights)


def load_partition(root_dir: str, partition_idx: int):
    """Returns (num_parts, graph, node_feat, edge_feat, node_pb, edge_pb);
    dict-valued for hetero datasets."""
    with open(os.path.join(root_dir, "META"), "rb") as f:
        meta = pickle.load(f)
    num_parts = meta["num_parts"]
    part_dir = os.path.join(root_dir, f"part{partition_idx}")
    if meta["data_cls"] == "homo":
        graph = _load_graph_dir(os.path.join(part_dir, "graph"))
        node_feat = _load_feature_dir(os.path.join(part_dir, "node_feat"))
        edge_feat = _load_feature_dir(os.path.join(part_dir, "edge_feat"))
        node_pb = torch.load(os.path.join(root_dir, "node_pb.pt"),
                             weights_only=False)
        edge_pb = torch.load(os.path.join(root_dir, "edge_pb.pt"),
                             weights_only=False)
        return num_parts, graph, node_feat, edge_feat, node_pb, edge_pb
    # hetero
    graph, node_feat, edge_feat = {}, {}, {}
    from ..typing import str2etype

    gdir = os.path.join(part_dir, "graph")
    for name in (os.listdir(gdir) if os.path.isdir(gdir) else []):
        graph[str2etype(name)] = _load_graph_dir(os.path.join(gdir, name))
    nfdir = os.path.join(part_dir, "node_feat")
    for name in (os.listdir(nfdir) if os.path.isdir(nfdir) else []):
        node_feat[name] = _load_feature_dir(os.path.join(nfdir, name))
    efdir = os.path.join(part_dir, "edge_feat")
    for name in (os.listdir(efdir) if os.path.isdir(efdir) else []):
        edge_feat[str2etype(name)] = _load_feature_dir(
            os.path.join(efdir, name))
    node_pb = {}
    for name in os.listdir(os.path.join(root_dir, "node_pb")):
        node_pb[name[:-3]] = torch.load(
            os.path.join(root_dir, "node_pb", name), weights_only=False)
    edge_pb = {}
    for name in os.listdir(os.path.join(root_dir, "edge_pb")):
        edge_pb[str2etype(name[:-3])] = torch.load(
            os.path.join(root_dir, "edge_pb", name), weights_only=False)
    return (num_parts, graph, node_feat or None, edge_feat or None, node_pb,
            edge_pb)


def cat_feature_cache(feat_part: FeaturePartitionData):
    """Prepend hot-cache rows to the partition's features; returns
    (feats, ids, id2index) where cache rows occupy positions [0, n_cache)
    (parity: reference partition/base.py:862-907)."""
    if feat_part.cache_feats is None:
        ids = feat_part.ids
        max_id = int(ids.max()) if ids.numel() else -1
        id2index = torch.full((max_id + 1,), -1, dtype=torch.long)
        id2index[ids] = torch.arange(ids.numel())
        return feat_part.feats, ids, id2index
    feats = torch.cat([feat_part.cache_feats, feat_part.feats])
    ids = torch.cat([feat_part.cache_ids, feat_part.ids])
    max_id = int(ids.max()) if ids.numel() else -1
    id2index = torch.full((max_id + 1,), -1, dtype=torch.long)
    # later (non-cache) entries must not override cache ids -> write feats
    # first, then cache positions win
    id2index[feat_part.ids] = torch.arange(feat_part.cache_ids.numel(),
                                           ids.numel())
    id2index[feat_part.cache_ids] = torch.arange(
        feat_part.cache_ids.numel())
    return feats, ids, id2index
