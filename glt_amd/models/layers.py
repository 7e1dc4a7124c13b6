"""GNN conv layers over glt_amd sampled batches.

The reference delegates modeling to PyG (reference README.md:279-303);
this image has no PyG, so glt_amd provides the conv layers its examples
and benchmarks need, written against the glt_amd batch convention:
``edge_index[0]`` = target-side (seed) local index, ``edge_index[1]`` =
source/neighbor local index; aggregation flows 1 -> 0.

Dense math lands on hipBLASLt through torch.nn.Linear (or the in-house
MFMA GEMM for skinny shapes); sparse aggregation uses the fused HIP
kernels in csrc/hip/ (segment-mean [+root concat], GAT edge-softmax) on
GPU fp32 batches, with an index_add_ fallback for CPU/other dtypes.
"""

import torch
import torch.nn as nn
import torch.nn.functional as F


def _degree(target: torch.Tensor, num_nodes: int) -> torch.Tensor:
    return torch.bincount(target, minlength=num_nodes).clamp_(min=1)


class SAGEConv(nn.Module):
    """GraphSAGE mean aggregator.

    The neighbor- and root-projections live in ONE [out, 2*in] parameter
    applied to [agg | x]: a single GEMM per layer and no per-step weight
    concatenation (the training step is launch-bound; see profiles/).
    """

    def __init__(self, in_channels: int, out_channels: int,
                 root_weight: bool = True, bias: bool = True):
        super().__init__()
        self.in_channels = in_channels
        self.root_weight = root_weight
        self.lin = nn.Linear(2 * in_channels if root_weight else
                             in_channels, out_channels, bias=bias)

    def forward(self, x, edge_index: torch.Tensor,
                num_target: int = None, sorted_by_target: bool = True,
                fuse_relu: bool = False) -> torch.Tensor:
        """x: [n, F] or (x_target, x_source) for bipartite relations.
        fuse_relu folds the activation into the projection GEMM epilogue
        (MFMA path); the caller must then skip its own activation."""
        if isinstance(x, tuple):
            x_tgt, x_src = x
            n = num_target if num_target is not None else x_tgt.size(0)
            tgt, src = edge_index[0], edge_index[1]
            if x_src.is_cuda and sorted_by_target and \
                    x_src.dtype in (torch.float32, torch.bfloat16):
                from ..ops import segment_mean

                agg = segment_mean(x_src, tgt, src, n)
            else:
                agg = x_src.new_zeros(n, x_src.size(1))
                agg.index_add_(0, tgt, x_src.index_select(0, src))
                agg = agg / _degree(tgt, n).unsqueeze(1).to(x_src.dtype)
            if self.root_weight:
                xin = torch.cat([agg, x_tgt[:n]], dim=1)
            else:
                xin = agg
            return self._project(xin, fuse_relu=False)
        n = num_target if num_target is not None else x.size(0)
        tgt, src = edge_index[0], edge_index[1]

        if (x.is_cuda and sorted_by_target
                and x.dtype in (torch.float32, torch.bfloat16)):
            # fused wave-per-row segment mean (glt_amd batches are sorted
            # by target local id by construction); with root_weight the
            # kernel also assembles [agg | x[:n]] in place of a dim-1 cat
            if self.root_weight:
                from ..ops import segment_mean_cat

                xin = segment_mean_cat(x, tgt, src, n)
            else:
                from ..ops import segment_mean

                xin = segment_mean(x, tgt, src, n)
        else:
            agg = x.new_zeros(n, x.size(1))
            agg.index_add_(0, tgt, x.index_select(0, src))
            agg = agg / _degree(tgt, n).unsqueeze(1).to(x.dtype)
            xin = torch.cat([agg, x[:n]], dim=1) if self.root_weight \
                else agg
        return self._project(xin, fuse_relu)

    def _project(self, xin, fuse_relu: bool):
        from ..ops import cast_linear, mfma_linear, use_mfma_linear

        if xin.dtype != self.lin.weight.dtype:
            # reduced-precision compute over fp32 master params (bf16
            # batches from a bf16 feature store)
            return cast_linear(xin, self.lin.weight, self.lin.bias,
                               relu=fuse_relu)
        if use_mfma_linear(xin, self.lin.weight, relu=fuse_relu):
            return mfma_linear(xin, self.lin.weight, self.lin.bias,
                               relu=fuse_relu)
        out = self.lin(xin)
        return F.relu(out) if fuse_relu else out


class GCNConv(nn.Module):
    """GCN with symmetric degree normalization (computed on the sampled
    subgraph)."""

    def __init__(self, in_channels: int, out_channels: int,
                 bias: bool = True):
        super().__init__()
        self.lin = nn.Linear(in_channels, out_channels, bias=bias)

    def forward(self, x, edge_index, num_target: int = None):
        n = x.size(0)
        nt = num_target if num_target is not None else n
        tgt, src = edge_index[0], edge_index[1]
        deg = _degree(torch.cat([tgt, src]), n).to(x.dtype)
        norm = deg.rsqrt()
        if x.dtype != self.lin.weight.dtype:
            from ..ops import cast_linear

            h = cast_linear(x, self.lin.weight, self.lin.bias)
        else:
            h = self.lin(x)
        msg = h.index_select(0, src) * norm[src].unsqueeze(1)
        out = h.new_zeros(nt, h.size(1))
        out.index_add_(0, tgt, msg)
        return out * norm[:nt].unsqueeze(1)


class GATConv(nn.Module):
    """Multi-head graph attention (GATv1-style, scatter softmax)."""

    def __init__(self, in_channels: int, out_channels: int, heads: int = 1,
                 concat: bool = True, negative_slope: float = 0.2,
                 dropout: float = 0.0, bias: bool = True):
        super().__init__()
        self.heads = heads
        self.out_channels = out_channels
        self.concat = concat
        self.negative_slope = negative_slope
        self.dropout = dropout
        self.lin = nn.Linear(in_channels, heads * out_channels, bias=False)
        self.att_src = nn.Parameter(torch.empty(1, heads, out_channels))
        self.att_dst = nn.Parameter(torch.empty(1, heads, out_channels))
        self.bias = nn.Parameter(torch.zeros(
            heads * out_channels if concat else out_channels)) if bias \
            else None
        nn.init.xavier_uniform_(self.att_src)
        nn.init.xavier_uniform_(self.att_dst)

    def forward(self, x, edge_index, num_target: int = None,
                sorted_by_target: bool = True):
        """x: [n, F] or a (x_target, x_source) tuple for bipartite edge
        sets (hetero relations); edge_index[0] indexes the target side,
        edge_index[1] the source side.  sorted_by_target: set False for
        edge sets not sorted by edge_index[0] — the fused segment kernel
        requires ascending targets; the scatter-softmax fallback does
        not."""
        def _proj(v):
            if v.dtype != self.lin.weight.dtype:
                from ..ops import cast_linear

                return cast_linear(v, self.lin.weight, self.lin.bias)
            return self.lin(v)

        if isinstance(x, tuple):
            x_tgt, x_src = x
            nt = num_target if num_target is not None else x_tgt.size(0)
            h_tgt = _proj(x_tgt).view(x_tgt.size(0), self.heads,
                                      self.out_channels)
            h_src = _proj(x_src).view(x_src.size(0), self.heads,
                                      self.out_channels)
        else:
            nt = num_target if num_target is not None else x.size(0)
            h_tgt = h_src = _proj(x).view(x.size(0), self.heads,
                                          self.out_channels)
        return self.attend(h_tgt, h_src, edge_index, nt,
                           sorted_by_target=sorted_by_target)

    def attend(self, h_tgt, h_src, edge_index, nt,
               sorted_by_target: bool = True):
        """Attention + aggregation over pre-projected features
        [n, heads, C] (lets HeteroConv batch the projections of all
        relations sharing a node type into one GEMM)."""
        tgt, src = edge_index[0], edge_index[1]
        h = h_src
        if (sorted_by_target and getattr(self, "use_fused", True)
                and h_src.is_cuda
                and h_src.dtype in (torch.float32, torch.bfloat16)
                and self.out_channels <= 128
                and not (self.training and self.dropout > 0)):
            # fused segment-softmax-aggregate (edges sorted by target);
            # the kernel computes the attention logits itself, so no
            # per-node alpha tensors are built here
            from ..ops import gat_softmax_aggregate

            out = gat_softmax_aggregate(h_tgt, h_src,
                                        self.att_src.squeeze(0),
                                        self.att_dst.squeeze(0),
                                        tgt, src, nt,
                                        self.negative_slope)
            out = out.reshape(nt, self.heads * self.out_channels) \
                if self.concat else out.mean(dim=1)
            if self.bias is not None:
                out = out + self.bias.to(out.dtype)
            return out
        if h_src.dtype != torch.float32:
            h_tgt = h_tgt.float()
            h_src = h = h_src.float()
        alpha_src = (h_src * self.att_src).sum(-1)
        alpha_dst = (h_tgt * self.att_dst).sum(-1)
        # index_select (not advanced indexing): its backward is an
        # index_add scatter, avoiding the radix-sort the indexing backward
        # performs per gather (24 device sorts/step in RGAT otherwise)
        e = alpha_dst.index_select(0, tgt) + alpha_src.index_select(0, src)
        e = F.leaky_relu(e, self.negative_slope)
        # scatter softmax over tgt
        e_max = torch.full((nt, self.heads), float("-inf"),
                           device=e.device, dtype=e.dtype)
        e_max.scatter_reduce_(0, tgt.unsqueeze(1).expand_as(e), e.detach(),
                              reduce="amax", include_self=True)
        # softmax is shift-invariant: the max is a constant offset, so it
        # carries no gradient (and its scatter-amax backward is skipped)
        e = (e - e_max.index_select(0, tgt)).exp()
        denom = torch.zeros(nt, self.heads, device=e.device, dtype=e.dtype)
        denom.index_add_(0, tgt, e)
        alpha = e / denom.clamp(min=1e-16).index_select(0, tgt)
        if self.training and self.dropout > 0:
            alpha = F.dropout(alpha, p=self.dropout)
        msg = h.index_select(0, src) * alpha.unsqueeze(-1)
        out = h.new_zeros(nt, self.heads, self.out_channels)
        out.index_add_(0, tgt, msg)
        out = out.reshape(nt, self.heads * self.out_channels) if self.concat \
            else out.mean(dim=1)
        if self.bias is not None:
            out = out + self.bias
        return out
