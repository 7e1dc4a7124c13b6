"""Misc utilities (parity: reference python/utils/common.py, tensor.py,
units.py, device.py, exit_status.py)."""
import os
import random
import socket
from contextlib import closing
from typing import Dict, List, Optional, Union

import torch

from ..typing import EdgeType, NodeType


def seed_everything(seed: int):
    """Seed python/torch and the native sampler RNG."""
    random.seed(seed)
    torch.manual_seed(seed)
    try:
        from .. import _C

        _C.manual_seed(seed)
    except ImportError:
        pass
    try:
        import numpy as np

        np.random.seed(seed % (2**32))
    except ImportError:
        pass


def tensor_equal_with_device(a: torch.Tensor, b: torch.Tensor) -> bool:
    """Equality including device placement (reference glt.utils helper)."""
    return a.device == b.device and a.shape == b.shape and bool(
        (a == b).all())


def id2idx(ids: Union[torch.Tensor, List[int]]) -> torch.Tensor:
    """Dense global-id -> position map (parity: utils/tensor.py:30-39)."""
    if not torch.is_tensor(ids):
        ids = torch.tensor(ids, dtype=torch.long)
    max_id = int(ids.max()) if ids.numel() > 0 else -1
    out = torch.zeros(max_id + 1, dtype=torch.long, device=ids.device)
    out[ids.long()] = torch.arange(ids.numel(), device=ids.device)
    return out


def index_select(data, index: torch.Tensor):
    if data is None:
        return None
    if isinstance(data, dict):
        return {k: index_select(v, index) for k, v in data.items()}
    return data[index.to(data.device if torch.is_tensor(data) else "cpu")]


def get_free_port(host: str = "127.0.0.1") -> int:
    with closing(socket.socket(socket.AF_INET, socket.SOCK_STREAM)) as s:
        s.bind((host, 0))
        s.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
        return s.getsockname()[1]


def parse_size(sz: Union[str, int, float, None]) -> Optional[int]:
    """'10GB' / '512MB' / int bytes -> bytes (parity: utils/units.py)."""
    if sz is None:
        return None
    if isinstance(sz, (int, float)):
        return int(sz)
    s = sz.strip().upper()
    units = {"KB": 2**10, "MB": 2**20, "GB": 2**30, "TB": 2**40, "B": 1,
             "K": 2**10, "M": 2**20, "G": 2**30, "T": 2**40}
    for u in ("KB", "MB", "GB", "TB", "K", "M", "G", "T", "B"):
        if s.endswith(u):
            return int(float(s[: -len(u)]) * units[u])
    return int(float(s))


def assign_device(rank: Optional[int] = None) -> torch.device:
    """Pick the local device for a worker (parity: utils/device.py:22-54)."""
    if torch.cuda.is_available():
        n = torch.cuda.device_count()
        idx = 0 if rank is None else rank % n
        return torch.device("cuda", idx)
    return torch.device("cpu")


def ensure_dir(path: str):
    os.makedirs(path, exist_ok=True)


def share_memory(t: Optional[torch.Tensor]):
    if t is not None and not t.is_cuda:
        t.share_memory_()
    return t


# -- checkpointing (parity: utils/common.py:177-234) -------------------------

def save_ckpt(ckpt_seq: int, ckpt_dir: str, model, optimizer=None,
              epoch: int = 0, keep: int = 5):
    ensure_dir(ckpt_dir)
    path = os.path.join(ckpt_dir, f"model_seq_{ckpt_seq}.ckpt")
    state = {
        "model": model.state_dict(),
        "optimizer": optimizer.state_dict() if optimizer is not None else None,
        "epoch": epoch,
        "seq": ckpt_seq,
    }
    tmp = path + ".tmp"
    torch.save(state, tmp)
    os.replace(tmp, path)
    # prune old checkpoints
    seqs = sorted(
        int(f[len("model_seq_"):-len(".ckpt")])
        for f in os.listdir(ckpt_dir)
        if f.startswith("model_seq_") and f.endswith(".ckpt")
    )
    for s in seqs[:-keep]:
        try:
            os.remove(os.path.join(ckpt_dir, f"model_seq_{s}.ckpt"))
        except OSError:
            pass
    return path


def load_ckpt(ckpt_dir: str, model, optimizer=None, seq: Optional[int] = None):
    """Load the latest (or given seq) checkpoint; returns epoch or -1."""
    if not os.path.isdir(ckpt_dir):
        return -1
    seqs = sorted(
        int(f[len("model_seq_"):-len(".ckpt")])
        for f in os.listdir(ckpt_dir)
        if f.startswith("model_seq_") and f.endswith(".ckpt")
    )
    if not seqs:
        return -1
    use = seq if seq is not None else seqs[-1]
    state = torch.load(os.path.join(ckpt_dir, f"model_seq_{use}.ckpt"),
                       map_location="cpu", weights_only=False)
    model.load_state_dict(state["model"])
    if optimizer is not None and state.get("optimizer") is not None:
        optimizer.load_state_dict(state["optimizer"])
    return state.get("epoch", -1)


# -- hetero sampler output helpers (parity: utils/common.py:85-135) ----------

def format_hetero_sampler_output(output, edge_dir: str = "out"):
    """Ensure every node type referenced by edges exists in output.node."""
    for etype in list(output.row.keys()):
        src, _, dst = etype
        for t in (src, dst):
            if t not in output.node:
                output.node[t] = torch.empty(
                    0, dtype=torch.long,
                    device=next(iter(output.node.values())).device
                    if output.node else None)
    return output


def merge_hetero_sampler_output(a, b, device=None, edge_dir: str = "out"):
    """Merge two HeteroSamplerOutput objects (concatenate + dedup nodes,
    concatenate relabeled edges).  Used when stitching multi-type link
    sampling results."""
    from ..sampler.base import HeteroSamplerOutput  # local import

    node = {}
    remap_a, remap_b = {}, {}
    types = set(a.node) | set(b.node)
    for t in types:
        na = a.node.get(t)
        nb = b.node.get(t)
        if na is None:
            node[t] = nb
            remap_b[t] = torch.arange(nb.numel(), device=nb.device)
            continue
        if nb is None:
            node[t] = na
            remap_a[t] = torch.arange(na.numel(), device=na.device)
            continue
        merged = torch.cat([na, nb])
        uniq, inv = torch.unique(merged, return_inverse=True, sorted=False)
        node[t] = uniq
        remap_a[t] = inv[: na.numel()]
        remap_b[t] = inv[na.numel():]

    row, col, edge = {}, {}, {}
    etypes = set(a.row) | set(b.row)
    for et in etypes:
        src, _, dst = et
        rs, cs = [], []
        es = []
        if et in a.row:
            rs.append(remap_a[src][a.row[et]] if src in remap_a else a.row[et])
            cs.append(remap_a[dst][a.col[et]] if dst in remap_a else a.col[et])
            if a.edge and et in a.edge:
                es.append(a.edge[et])
        if et in b.row:
            rs.append(remap_b[src][b.row[et]] if src in remap_b else b.row[et])
            cs.append(remap_b[dst][b.col[et]] if dst in remap_b else b.col[et])
            if b.edge and et in b.edge:
                es.append(b.edge[et])
        row[et] = torch.cat(rs)
        col[et] = torch.cat(cs)
        if es:
            edge[et] = torch.cat(es)
    return HeteroSamplerOutput(
        node=node, row=row, col=col, edge=edge or None,
        edge_types=list(etypes), device=device or a.device,
        metadata=a.metadata)
