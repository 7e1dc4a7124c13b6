"""Online distributed random partitioner (capability parity: reference
python/distributed/dist_random_partitioner.py:61-539).

Every rank holds a chunk of the global graph (nodes/edges/features).  The
node -> partition assignment is a deterministic keyed hash shared by all
ranks (no broadcast needed); each rank then pushes the rows/edges it holds
to their owning partitions over RPC and receives its own.  The result is
the same per-partition data `load_partition` would produce, built in
memory (optionally saved in the GLT on-disk layout).
"""
import threading
from typing import Dict, List, Optional, Tuple

import torch

from ..partition import GLTPartitionBook
from ..partition.base import (FeaturePartitionData, GraphPartitionData,
                              save_edge_pb, save_graph_partition,
                              save_feature_partition, save_meta,
                              save_node_pb)
from ..utils.common import ensure_dir
from .dist_context import get_context
from .rpc import RpcCalleeBase, barrier, rpc_register, rpc_request_async


class _PartitionReceiver(RpcCalleeBase):
    def __init__(self):
        self.lock = threading.Lock()
        self.edge_chunks: List[torch.Tensor] = []
        self.eid_chunks: List[torch.Tensor] = []
        self.feat_chunks: List[torch.Tensor] = []
        self.feat_id_chunks: List[torch.Tensor] = []

    def call(self, kind: str, *tensors):
        with self.lock:
            if kind == "edges":
                ei, eids = tensors
                self.edge_chunks.append(ei)
                self.eid_chunks.append(eids)
            elif kind == "feats":
                feats, ids = tensors
                self.feat_chunks.append(feats)
                self.feat_id_chunks.append(ids)
        return True


class DistRandomPartitioner:
    """Args:
      num_nodes: global node count.
      local_edge_index: [2, E_local] the edges this rank holds (global ids).
      local_eids: their global edge ids (default: inferred offset range).
      local_node_feat/local_node_ids: feature rows this rank holds.
      edge_assign_strategy: 'by_src' | 'by_dst'.
      seed: shared hash seed (must match across ranks).
    """

    def __init__(self, num_nodes: int, local_edge_index: torch.Tensor,
                 local_eids: Optional[torch.Tensor] = None,
                 local_node_feat: Optional[torch.Tensor] = None,
                 local_node_ids: Optional[torch.Tensor] = None,
                 edge_assign_strategy: str = "by_src", seed: int = 0):
        ctx = get_context()
        assert ctx is not None, "init_worker_group + init_rpc first"
        self.ctx = ctx
        self.num_parts = ctx.world_size
        self.rank = ctx.rank
        self.num_nodes = num_nodes
        self.local_edge_index = local_edge_index
        self.local_eids = local_eids
        self.local_node_feat = local_node_feat
        self.local_node_ids = local_node_ids
        self.edge_assign_strategy = edge_assign_strategy
        self.seed = seed
        self._receiver = _PartitionReceiver()
        self._callee_id = rpc_register(self._receiver)

    def node_pb(self) -> GLTPartitionBook:
        """Deterministic shared assignment: permutation-free keyed hash."""
        g = torch.Generator()
        g.manual_seed(self.seed)
        book = torch.randint(0, self.num_parts, (self.num_nodes,),
                             generator=g, dtype=torch.uint8)
        return GLTPartitionBook(book)

    def partition(self) -> Tuple[GLTPartitionBook, GraphPartitionData,
                                 Optional[FeaturePartitionData]]:
        barrier()
        pb = self.node_pb()
        book = pb.book.long()
        ei = self.local_edge_index
        eids = self.local_eids
        if eids is None:
            eids = torch.arange(ei.size(1))
        assign = ei[0] if self.edge_assign_strategy == "by_src" else ei[1]
        owner = book[assign]
        futs = []
        for p in range(self.num_parts):
            mask = owner == p
            if not bool(mask.any()):
                continue
            chunk_ei = ei[:, mask]
            chunk_eids = eids[mask]
            if p == self.rank:
                self._receiver.call("edges", chunk_ei, chunk_eids)
            else:
                futs.append(rpc_request_async(
                    p, self._callee_id, args=("edges", chunk_ei,
                                              chunk_eids)))
        if self.local_node_feat is not None:
            ids = self.local_node_ids
            if ids is None:
                ids = torch.arange(self.local_node_feat.size(0))
            owners = book[ids]
            for p in range(self.num_parts):
                mask = owners == p
                if not bool(mask.any()):
                    continue
                if p == self.rank:
                    self._receiver.call("feats",
                                        self.local_node_feat[mask],
                                        ids[mask])
                else:
                    futs.append(rpc_request_async(
                        p, self._callee_id,
                        args=("feats", self.local_node_feat[mask],
                              ids[mask])))
        for f in futs:
            f.wait()
        barrier()  # all pushes delivered
        r = self._receiver
        graph = GraphPartitionData(
            edge_index=torch.cat(r.edge_chunks, dim=1)
            if r.edge_chunks else torch.empty(2, 0, dtype=torch.long),
            eids=torch.cat(r.eid_chunks)
            if r.eid_chunks else torch.empty(0, dtype=torch.long))
        feat = None
        if r.feat_chunks:
            ids = torch.cat(r.feat_id_chunks)
            order = torch.argsort(ids)
            feat = FeaturePartitionData(
                feats=torch.cat(r.feat_chunks)[order], ids=ids[order])
        return pb, graph, feat

    def partition_and_save(self, output_dir: str):
        pb, graph, feat = self.partition()
        ensure_dir(output_dir)
        if self.rank == 0:
            save_meta(output_dir, self.num_parts)
            save_node_pb(output_dir, pb)
            save_edge_pb(output_dir, pb)  # edges assigned by node owner
        save_graph_partition(output_dir, self.rank, graph)
        if feat is not None:
            save_feature_partition(output_dir, self.rank, feat)
        barrier()
        return pb, graph, feat


class DistHeteroRandomPartitioner:
    """Hetero online partitioning (capability parity: reference
    dist_random_partitioner.py:300-539): one deterministic node book per
    node TYPE (so every edge type sharing a src type agrees on ownership),
    per-edge-type edge exchange and per-node-type feature exchange, all
    composed from the homo partitioner's RPC push machinery.

    Every rank must construct with the same type sets (callee
    registration and barriers run in sorted-type order).
    """

    def __init__(self, num_nodes: Dict[str, int],
                 local_edge_index: Dict[tuple, torch.Tensor],
                 local_eids: Optional[Dict[tuple, torch.Tensor]] = None,
                 local_node_feat: Optional[Dict[str, torch.Tensor]] = None,
                 local_node_ids: Optional[Dict[str, torch.Tensor]] = None,
                 edge_assign_strategy: str = "by_src", seed: int = 0):
        import zlib

        self.edge_types = sorted(local_edge_index.keys())
        self.node_types = sorted(num_nodes.keys())
        self.edge_assign_strategy = edge_assign_strategy

        def type_seed(nt: str) -> int:
            return seed * 0x9E3779B9 + zlib.crc32(nt.encode())

        # per-edge-type partitioners exchange edges; the book each uses is
        # the ASSIGN-side node type's book (same seed per type across all
        # edge types -> consistent ownership)
        self._edge_parts = {}
        for et in self.edge_types:
            assign_t = et[0] if edge_assign_strategy == "by_src" else et[2]
            self._edge_parts[et] = DistRandomPartitioner(
                num_nodes[assign_t], local_edge_index[et],
                local_eids.get(et) if local_eids else None,
                edge_assign_strategy=edge_assign_strategy,
                seed=type_seed(assign_t))
        # per-node-type partitioners exchange feature rows
        self._feat_parts = {}
        for nt in self.node_types:
            feat = local_node_feat.get(nt) if local_node_feat else None
            if feat is None:
                continue
            self._feat_parts[nt] = DistRandomPartitioner(
                num_nodes[nt],
                torch.empty(2, 0, dtype=torch.long),
                local_node_feat=feat,
                local_node_ids=(local_node_ids.get(nt)
                                if local_node_ids else None),
                seed=type_seed(nt))
        self._num_nodes = num_nodes
        self._seed_of = type_seed

    def node_pbs(self) -> Dict[str, GLTPartitionBook]:
        out = {}
        for nt in self.node_types:
            helper = DistRandomPartitioner.__new__(DistRandomPartitioner)
            helper.seed = self._seed_of(nt)
            helper.num_nodes = self._num_nodes[nt]
            helper.num_parts = get_context().world_size
            out[nt] = DistRandomPartitioner.node_pb(helper)
        return out

    def partition(self):
        """Returns (node_pbs, graph_parts, feat_parts) — all dicts keyed
        by node/edge type; this rank's shard of each."""
        graphs, feats = {}, {}
        for et in self.edge_types:  # same order on every rank
            _, graphs[et], _ = self._edge_parts[et].partition()
        for nt in sorted(self._feat_parts.keys()):
            _, _, feats[nt] = self._feat_parts[nt].partition()
        return self.node_pbs(), graphs, feats
