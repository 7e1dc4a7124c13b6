"""SampleMessage encoding: SamplerOutput(+collected features/labels) <->
flat Dict[str, Tensor] for shm/RPC transport.

Key conventions (capability parity with the reference's '#IS_HETERO' /
'#META.*' flat dicts, reference dist_neighbor_sampler.py:689-807):
  '#IH'                     [1] uint8: 1 if hetero
  '#END'                    end-of-epoch marker
  homo:   node,row,col,edge?,batch?,nsn,nse,x?,y?,edge_attr?
  hetero: n.<ntype>.{node,batch,nsn,x?,y?}  e.<etype str>.{row,col,edge?,
          nse,edge_attr?}
  metadata tensors: m.<key>;  string metadata: ms.<key> (utf-8 bytes)
"""
from typing import Dict, Optional, Tuple, Union

import torch

from ..sampler import HeteroSamplerOutput, SamplerOutput
from ..typing import as_str, str2etype

END_KEY = "#END"


def _s2t(s: str) -> torch.Tensor:
    return torch.frombuffer(bytearray(s.encode("utf-8")),
                            dtype=torch.uint8).clone()


def _t2s(t: torch.Tensor) -> str:
    return bytes(t.tolist()).decode("utf-8")


def _put_meta(msg, md):
    if not isinstance(md, dict):
        return
    for k, v in md.items():
        if torch.is_tensor(v):
            msg[f"m.{k}"] = v
        elif isinstance(v, str):
            msg[f"ms.{k}"] = _s2t(v)
        elif isinstance(v, tuple) and all(isinstance(x, str) for x in v):
            msg[f"ms.{k}"] = _s2t(as_str(v))


def _get_meta(msg) -> dict:
    md = {}
    for k, v in msg.items():
        if k.startswith("m."):
            md[k[2:]] = v
        elif k.startswith("ms."):
            s = _t2s(v)
            md[k[3:]] = str2etype(s)
    return md


def encode_sampler_output(
        out: Union[SamplerOutput, HeteroSamplerOutput],
        x=None, y=None, edge_attr=None) -> Dict[str, torch.Tensor]:
    msg: Dict[str, torch.Tensor] = {}
    if isinstance(out, SamplerOutput):
        msg["#IH"] = torch.zeros(1, dtype=torch.uint8)
        msg["node"] = out.node
        msg["row"] = out.row
        msg["col"] = out.col
        if out.edge is not None:
            msg["edge"] = out.edge
        if out.batch is not None:
            msg["batch"] = out.batch
        if out.num_sampled_nodes is not None:
            msg["nsn"] = torch.as_tensor(out.num_sampled_nodes)
        if out.num_sampled_edges is not None:
            msg["nse"] = torch.as_tensor(out.num_sampled_edges)
        if x is not None:
            msg["x"] = x
        if y is not None:
            msg["y"] = y
        if edge_attr is not None:
            msg["edge_attr"] = edge_attr
        _put_meta(msg, out.metadata)
        return msg

    msg["#IH"] = torch.ones(1, dtype=torch.uint8)
    for nt, nodes in out.node.items():
        msg[f"n.{nt}.node"] = nodes
        if out.batch and nt in out.batch:
            msg[f"n.{nt}.batch"] = out.batch[nt]
        if out.num_sampled_nodes and nt in out.num_sampled_nodes:
            msg[f"n.{nt}.nsn"] = torch.as_tensor(out.num_sampled_nodes[nt])
        if x and nt in x:
            msg[f"n.{nt}.x"] = x[nt]
        if y and nt in y:
            msg[f"n.{nt}.y"] = y[nt]
    for et, r in out.row.items():
        es = as_str(et)
        msg[f"e.{es}.row"] = r
        msg[f"e.{es}.col"] = out.col[et]
        if out.edge and et in out.edge:
            msg[f"e.{es}.edge"] = out.edge[et]
        if out.num_sampled_edges and et in out.num_sampled_edges:
            msg[f"e.{es}.nse"] = torch.as_tensor(out.num_sampled_edges[et])
        if edge_attr and et in edge_attr:
            msg[f"e.{es}.edge_attr"] = edge_attr[et]
    if isinstance(out.input_type, (tuple, str)) and out.input_type:
        msg["ms.input_type"] = _s2t(as_str(out.input_type))
    _put_meta(msg, out.metadata)
    return msg


def decode_sample_message(msg: Dict[str, torch.Tensor]):
    """Returns (sampler_output, x, y, edge_attr) — dict-valued for hetero."""
    md = _get_meta(msg)
    if int(msg["#IH"].item()) == 0:
        out = SamplerOutput(
            node=msg["node"], row=msg["row"], col=msg["col"],
            edge=msg.get("edge"), batch=msg.get("batch"),
            num_sampled_nodes=(msg["nsn"].tolist()
                               if "nsn" in msg else None),
            num_sampled_edges=(msg["nse"].tolist()
                               if "nse" in msg else None),
            metadata=md or None)
        return out, msg.get("x"), msg.get("y"), msg.get("edge_attr")
    node, batch, nsn, x, y = {}, {}, {}, {}, {}
    row, col, edge, nse, ea = {}, {}, {}, {}, {}
    for k, v in msg.items():
        if k.startswith("n."):
            _, nt, field = k.split(".", 2)
            if field == "node":
                node[nt] = v
            elif field == "batch":
                batch[nt] = v
            elif field == "nsn":
                nsn[nt] = v.tolist()
            elif field == "x":
                x[nt] = v
            elif field == "y":
                y[nt] = v
        elif k.startswith("e."):
            _, es, field = k.split(".", 2)
            et = str2etype(es)
            if field == "row":
                row[et] = v
            elif field == "col":
                col[et] = v
            elif field == "edge":
                edge[et] = v
            elif field == "nse":
                nse[et] = v.tolist()
            elif field == "edge_attr":
                ea[et] = v
    input_type = md.pop("input_type", None)
    out = HeteroSamplerOutput(
        node=node, row=row, col=col, edge=edge or None,
        batch=batch or None, num_sampled_nodes=nsn or None,
        num_sampled_edges=nse or None, input_type=input_type,
        metadata=md or None)
    return out, x or None, y or None, ea or None
