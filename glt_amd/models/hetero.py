"""Heterogeneous models: RGNN (rgat / rsage, as in the reference MLPerf
IGBH example, reference examples/igbh/rgnn.py capability) and a generic
HeteroConv combinator."""
from typing import Dict, List, Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..typing import EdgeType, NodeType
from .layers import GATConv, SAGEConv


class HeteroConv(nn.Module):
    """Applies a per-edge-type conv and sums contributions per dst type."""

    def __init__(self, convs: Dict[EdgeType, nn.Module], aggr: str = "sum"):
        super().__init__()
        self.convs = nn.ModuleDict({"__".join(k): v for k, v in convs.items()})
        self.aggr = aggr

    def forward(self, x_dict: Dict[NodeType, torch.Tensor],
                edge_index_dict: Dict[EdgeType, torch.Tensor],
                self_lins: Optional[Dict[NodeType, nn.Module]] = None):
        """With `self_lins` (a per-node-type Linear dict, e.g. RGNN's self
        projections), returns (conv_out, self_out); the self projections
        ride along in the batched per-type GEMM where shapes allow."""
        rels = []
        for etype, ei in edge_index_dict.items():
            key = "__".join(etype)
            if key not in self.convs:
                continue
            # target side of a glt_amd batch edge is etype-dependent: the
            # sampler emits edges keyed so that edge_index[0] is the
            # walked-from (seed-side) type.
            src_t, _, dst_t = etype
            if x_dict.get(src_t) is None or x_dict.get(dst_t) is None:
                continue
            rels.append((etype, key, ei))
        self_out = {} if self_lins is not None else None
        if rels and all(isinstance(self.convs[k], GATConv)
                        for _, k, _ in rels):
            out = self._multi_gat(x_dict, rels, self_lins, self_out) \
                if getattr(self, "use_multi", True) else None
            if out is None:
                out = self._batched_gat(x_dict, rels, self_lins,
                                        self_out)
        else:
            out = {}
            for etype, key, ei in rels:
                src_t, _, dst_t = etype
                x_tgt, x_src = x_dict[src_t], x_dict[dst_t]
                conv = self.convs[key]
                try:
                    h = conv((x_tgt, x_src), ei)
                except TypeError:
                    h = conv_bipartite(conv, x_tgt, x_src, ei)
                out.setdefault(src_t, []).append(h)
        if self_out is not None:
            for t, x in x_dict.items():
                if t not in self_out and t in self_lins:
                    lin = self_lins[t]
                    if x.dtype != lin.weight.dtype:
                        from ..ops import cast_linear

                        self_out[t] = cast_linear(x, lin.weight, lin.bias)
                    else:
                        self_out[t] = lin(x)
        result = {}
        for t, hs in out.items():
            acc = hs[0]
            for v in hs[1:]:  # chained adds: no stack copy, same grads
                acc = acc + v
            result[t] = acc
        return result if self_out is None else (result, self_out)


    def _multi_gat(self, x_dict, rels, self_lins=None, self_out=None):
        """All relations' attention in ONE fused kernel launch per layer
        (RGAT is launch-bound: ~285 kernels/step measured round 2).  The
        per-type batched projection is consumed in place via strided
        views and the backward accumulates one dh arena per type, so the
        per-relation .contiguous() copies, slice-grad zeros and adds all
        disappear.  Returns None when preconditions fail (mixed heads /
        dims, CPU, dropout-in-training, >8 relations) so the caller
        falls back to the per-relation path."""
        if len(rels) > 8:
            return None
        convs = {k: self.convs[k] for _, k, _ in rels}
        cs = list(convs.values())
        c0 = cs[0]
        if any(c.heads != c0.heads or c.out_channels != c0.out_channels
               or not getattr(c, "use_fused", True)
               or (self.training and c.dropout > 0)
               or c.negative_slope != c0.negative_slope
               or c.lin.bias is not None for c in cs):
            return None
        if c0.out_channels > 128:
            return None
        xs = [x_dict[t] for t in {t for et, _, _ in rels
                                  for t in (et[0], et[2])}]
        if not all(x.is_cuda and x.dtype in (torch.float32,
                                             torch.bfloat16)
                   and x.dtype == xs[0].dtype for x in xs):
            return None
        from ..ops import cast_linear
        from ..ops.gat import gat_multi_layer
        from ..ops.segment import _boundaries

        heads, C = c0.heads, c0.out_channels
        needs: Dict[NodeType, List[str]] = {}
        for etype, key, _ in rels:
            for t in (etype[0], etype[2]):
                ks = needs.setdefault(t, [])
                if key not in ks:
                    ks.append(key)
        types = list(needs.keys())
        tidx = {t: i for i, t in enumerate(types)}
        h_list: List[Optional[torch.Tensor]] = [None] * len(types)
        col_off = {}
        for t, keys in needs.items():
            x = x_dict[t]
            ws = [convs[k].lin.weight for k in keys]
            if len({w.size(1) for w in ws}) != 1 or \
                    ws[0].size(1) != x.size(1):
                return None
            fold_self = (self_lins is not None and t in self_lins
                         and self_lins[t].weight.size(1) == x.size(1))
            if fold_self:
                ws = ws + [self_lins[t].weight]
            W = ws[0] if len(ws) == 1 else torch.cat(ws, dim=0)
            if x.dtype != W.dtype:
                h = cast_linear(x, W, None)
            else:
                h = torch.nn.functional.linear(x, W)
            off = 0
            for k in keys:
                col_off[(t, k)] = off
                off += heads * C
            if fold_self:
                sl = h[:, off:]
                b = self_lins[t].bias
                self_out[t] = sl if b is None else sl + b.to(sl.dtype)
            h_list[tidx[t]] = h
        specs, asl, adl, bl, srcs, offs = [], [], [], [], [], []
        rel_meta = []
        for etype, key, ei in rels:
            src_t, _, dst_t = etype
            conv = convs[key]
            nt = x_dict[src_t].size(0)
            tgt = ei[0]
            off_r = torch.searchsorted(tgt, _boundaries(nt, tgt.device))
            specs.append((tidx[src_t], col_off[(src_t, key)],
                          tidx[dst_t], col_off[(dst_t, key)]))
            asl.append(conv.att_src)
            adl.append(conv.att_dst)
            # bias folded into the kernel epilogue; for concat=False the
            # per-head repeat commutes with the head mean, and repeat's
            # backward folds the [H*C] kernel grad back to [C]
            if conv.bias is None:
                bl.append(torch.empty(0, device=tgt.device))
            elif conv.concat:
                bl.append(conv.bias)
            else:
                bl.append(conv.bias.repeat(heads))
            srcs.append(ei[1].contiguous())
            offs.append(off_r)
            rel_meta.append((etype, key, nt))
        spec = {"H": heads, "C": C, "rels": specs}
        arena = gat_multi_layer(c0.negative_slope, spec, asl, adl, bl,
                                h_list, srcs, offs)
        out: Dict[NodeType, List[torch.Tensor]] = {}
        parts = torch.split(arena, [nt for _, _, nt in rel_meta], dim=0)
        for (etype, key, nt), part in zip(rel_meta, parts):
            conv = convs[key]
            o = part.reshape(nt, heads * C) if conv.concat \
                else part.mean(dim=1)
            out.setdefault(etype[0], []).append(o)
        return out

    def _batched_gat(self, x_dict, rels, self_lins=None, self_out=None):
        """All-GAT layers: batch the per-relation projections of each node
        type into ONE GEMM (RGAT spends most of its host+GEMM time on many
        small per-relation Linears otherwise), then run each relation's
        attention over views of the stacked output.  When a matching
        per-type self projection is supplied, its weight rides in the same
        GEMM (bias added on the slice)."""
        needs: Dict[NodeType, List[str]] = {}
        for etype, key, _ in rels:
            for t in (etype[0], etype[2]):
                keys = needs.setdefault(t, [])
                if key not in keys:
                    keys.append(key)
        H = {}
        for t, keys in needs.items():
            x = x_dict[t]
            convs = [self.convs[k] for k in keys]
            in_dims = {c.lin.weight.size(1) for c in convs}
            if len(in_dims) > 1:
                for k, c in zip(keys, convs):
                    if x.dtype != c.lin.weight.dtype:
                        from ..ops import cast_linear

                        hx = cast_linear(x, c.lin.weight, c.lin.bias)
                    else:
                        hx = c.lin(x)
                    H[(t, k)] = hx.view(x.size(0), c.heads,
                                        c.out_channels)
                continue
            weights = [c.lin.weight for c in convs]
            fold_self = (self_lins is not None and t in self_lins
                         and self_lins[t].weight.size(1) == x.size(1))
            if fold_self:
                weights.append(self_lins[t].weight)
            W = weights[0] if len(weights) == 1 \
                else torch.cat(weights, dim=0)
            if x.dtype != W.dtype:
                from ..ops import cast_linear

                h = cast_linear(x, W, None)
            else:
                h = torch.nn.functional.linear(x, W)
            # torch.split: the backward is ONE cat of the slice grads —
            # manual h[:, a:b] views made autograd materialize zeros(h)
            # + add per relation (a large share of RGAT's 49 fills and
            # 20 adds per step; see profiles/r02 RGAT notes)
            sizes = [c.heads * c.out_channels for c in convs]
            if fold_self:
                sizes.append(h.size(1) - sum(sizes))
            parts = torch.split(h, sizes, dim=1) if len(sizes) > 1 \
                else (h,)
            for k, c, hp in zip(keys, convs, parts):
                H[(t, k)] = hp.contiguous().view(
                    x.size(0), c.heads, c.out_channels)
            if fold_self:
                s = parts[-1]
                b = self_lins[t].bias
                self_out[t] = s if b is None else s + b.to(s.dtype)
        out: Dict[NodeType, List[torch.Tensor]] = {}
        for etype, key, ei in rels:
            src_t, _, dst_t = etype
            conv = self.convs[key]
            h = conv.attend(H[(src_t, key)], H[(dst_t, key)], ei,
                            x_dict[src_t].size(0))
            out.setdefault(src_t, []).append(h)
        return out


def conv_bipartite(conv, x_tgt, x_src, edge_index):
    """Run a homogeneous conv on a bipartite edge set by stacking target
    and source feature rows into one local space."""
    n_tgt = x_tgt.size(0)
    x = torch.cat([x_tgt, x_src], dim=0)
    ei = torch.stack([edge_index[0], edge_index[1] + n_tgt])
    return conv(x, ei)[:n_tgt]


class RGNN(nn.Module):
    """Relational GNN over hetero batches; model='rgat' or 'rsage'.

    Per layer: per-edge-type conv (GAT or SAGE) summed per target node
    type, plus a per-node-type self projection so every type advances to
    the layer's output dim even when it receives no messages; ReLU/dropout
    between layers; linear head on the predicted node type.
    """

    def __init__(self, etypes: List[EdgeType], in_dim: int, h_dim: int,
                 out_dim: int, num_layers: int = 2, n_heads: int = 4,
                 model: str = "rgat", dropout: float = 0.2,
                 node_types: Optional[List[NodeType]] = None):
        super().__init__()
        self.model = model
        if node_types is None:
            node_types = sorted({t for et in etypes for t in (et[0], et[2])})
        self.node_types = node_types
        self.layers = nn.ModuleList()
        self.self_lins = nn.ModuleList()
        dims = [in_dim] + [h_dim] * num_layers
        for li in range(num_layers):
            convs = {}
            for et in etypes:
                if model == "rgat":
                    convs[et] = GATConv(dims[li], dims[li + 1] // n_heads,
                                        heads=n_heads, concat=True)
                else:
                    convs[et] = SAGEConv(dims[li], dims[li + 1],
                                         root_weight=False)
            self.layers.append(HeteroConv(convs))
            self.self_lins.append(nn.ModuleDict(
                {t: nn.Linear(dims[li], dims[li + 1])
                 for t in node_types}))
        self.head = nn.Linear(h_dim, out_dim)
        self.dropout = dropout

    def forward(self, x_dict: Dict[NodeType, torch.Tensor],
                edge_index_dict: Dict[EdgeType, torch.Tensor],
                predict_type: Optional[NodeType] = None):
        h = x_dict
        for i, layer in enumerate(self.layers):
            h_conv, self_out = layer(h, edge_index_dict,
                                     self_lins=self.self_lins[i])
            h_next = {}
            for t, v in h.items():
                if t not in self.self_lins[i]:
                    continue
                out = self_out[t]
                hc = h_conv.get(t)
                if hc is not None:
                    # conv output may cover fewer rows than the projection
                    # (message targets only); add on the common prefix
                    m = min(hc.size(0), out.size(0))
                    if m == out.size(0) and m == hc.size(0):
                        out = out + hc  # full cover: skip the cat copy
                    else:
                        out = torch.cat([out[:m] + hc[:m], out[m:]], dim=0)
                h_next[t] = F.dropout(F.relu(out), p=self.dropout,
                                      training=self.training)
            h = h_next
        def _head(v):
            if v.dtype != self.head.weight.dtype:
                from ..ops import cast_linear

                return cast_linear(v, self.head.weight, self.head.bias)
            return self.head(v)

        if predict_type is not None:
            return _head(h[predict_type])
        return {t: _head(v) for t, v in h.items()}
