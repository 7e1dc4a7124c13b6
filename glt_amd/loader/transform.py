"""SamplerOutput -> Data / HeteroData (parity: reference
python/loader/transform.py:26-136)."""
from typing import Dict, Literal, Optional

import torch

from ..pygcompat import Data, HeteroData
from ..sampler import HeteroSamplerOutput, SamplerOutput
from ..typing import EdgeType, NodeType, reverse_edge_type


def to_data(sampler_out: SamplerOutput,
            batch_labels: Optional[torch.Tensor] = None,
            node_feats: Optional[torch.Tensor] = None,
            edge_feats: Optional[torch.Tensor] = None, **kwargs) -> Data:
    edge_index = torch.stack([sampler_out.row, sampler_out.col])
    data = Data(x=node_feats, edge_index=edge_index, y=batch_labels, **kwargs)
    if edge_feats is not None:
        data.edge_attr = edge_feats
    data.edge = sampler_out.edge
    data.node = sampler_out.node
    data.batch = sampler_out.batch
    data.batch_size = (sampler_out.batch.numel()
                       if sampler_out.batch is not None else 0)
    data.num_sampled_nodes = sampler_out.num_sampled_nodes
    data.num_sampled_edges = sampler_out.num_sampled_edges
    md = sampler_out.metadata
    if isinstance(md, dict):
        for k, v in md.items():
            if k == "edge_label_index" and v is not None:
                # binary link prediction: flip into message-flow direction
                data["edge_label_index"] = torch.stack((v[1], v[0]))
            elif k != "input_type":
                data[k] = v
    elif md is not None:
        data["metadata"] = md
    return data


def to_hetero_data(out: HeteroSamplerOutput,
                   batch_label_dict: Optional[Dict[NodeType,
                                                   torch.Tensor]] = None,
                   node_feat_dict: Optional[Dict[NodeType,
                                                 torch.Tensor]] = None,
                   edge_feat_dict: Optional[Dict[EdgeType,
                                                 torch.Tensor]] = None,
                   edge_dir: Literal["in", "out"] = "out",
                   **kwargs) -> HeteroData:
    data = HeteroData()
    for k, v in kwargs.items():
        setattr(data, k, v)

    num_hops = max((len(v) for v in (out.num_sampled_edges or {}).values()),
                   default=0)

    for et, r in out.row.items():
        data[et].edge_index = torch.stack([r, out.col[et]])
        if out.edge is not None and et in out.edge:
            data[et].edge = out.edge[et]
        if edge_feat_dict is not None and et in edge_feat_dict:
            data[et].edge_attr = edge_feat_dict[et]
        ne = (out.num_sampled_edges or {}).get(et, [])
        data[et].num_sampled_edges = list(ne) + [0] * (num_hops - len(ne))

    for nt, nodes in out.node.items():
        data[nt].node = nodes
        if node_feat_dict is not None and nt in node_feat_dict:
            data[nt].x = node_feat_dict[nt]
        nn = (out.num_sampled_nodes or {}).get(nt, [])
        data[nt].num_sampled_nodes = list(nn) + [0] * (num_hops + 1 - len(nn))

    for nt, b in (out.batch or {}).items():
        data[nt].batch = b
        data[nt].batch_size = b.numel()
        if batch_label_dict is not None and nt in batch_label_dict:
            data[nt].y = batch_label_dict[nt]

    md = out.metadata
    input_type = out.input_type
    if isinstance(md, dict):
        res_et = (reverse_edge_type(input_type)
                  if (edge_dir == "out" and isinstance(input_type, tuple))
                  else input_type)
        for k, v in md.items():
            if v is None or k in ("input_type", "bs"):
                continue
            if k == "edge_label_index":
                data[res_et]["edge_label_index"] = (
                    torch.stack((v[1], v[0])) if edge_dir == "out" else v)
            elif k == "edge_label":
                data[res_et]["edge_label"] = v
            elif k == "src_index":
                data[input_type[0]]["src_index"] = v
            elif k in ("dst_pos_index", "dst_neg_index"):
                data[input_type[-1]][k] = v
            else:
                setattr(data, k, v)
    elif md is not None:
        data.metadata = md
    return data
