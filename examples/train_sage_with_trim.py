#!/usr/bin/env python3
"""Importance-trimmed training: compute per-node inclusion probabilities
with `NeighborSampler.sample_prob` (the CalNbrProb propagation), use them to
frequency-partition features / pick a GPU hot cache, then train GraphSAGE
(capability parity: reference examples/train_sage_prod_with_trim.py +
FrequencyPartitioner flow)."""
import argparse
import os
import sys
import tempfile
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.nn.functional as F

import glt_amd
from glt_amd import Dataset, NeighborLoader
from glt_amd.data import Feature, Graph, Topology, sort_by_in_degree
from glt_amd.models import GraphSAGE
from glt_amd.partition import FrequencyPartitioner, load_partition
from glt_amd.sampler import NeighborSampler


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--nodes", type=int, default=200_000)
    ap.add_argument("--edges", type=int, default=2_000_000)
    ap.add_argument("--feat-dim", type=int, default=64)
    ap.add_argument("--fanout", type=str, default="10,5")
    ap.add_argument("--batch-size", type=int, default=512)
    ap.add_argument("--cache-ratio", type=float, default=0.2)
    ap.add_argument("--steps", type=int, default=50)
    args = ap.parse_args()

    has_gpu = torch.cuda.is_available()
    if not has_gpu:
        args.nodes, args.edges, args.steps = 5_000, 50_000, 10
    device = torch.device("cuda", 0) if has_gpu else torch.device("cpu")
    glt_amd.seed_everything(0)
    n = args.nodes
    src = torch.randint(0, n, (args.edges,))
    dst = torch.randint(0, n, (args.edges,))
    edge_index = torch.stack([src, dst])
    feats = torch.randn(n, args.feat_dim)
    labels = torch.randint(0, 10, (n,))
    fanout = [int(x) for x in args.fanout.split(",")]

    # 1) probability propagation from a training-seed sample
    topo = Topology(edge_index, num_nodes=n)
    g = Graph(topo, mode="CUDA" if has_gpu else "CPU",
              device=0 if has_gpu else None)
    sampler = NeighborSampler(g, fanout)
    seeds = torch.randperm(n)[: n // 10]
    probs = sampler.sample_prob(seeds.to(device) if has_gpu else seeds, n)
    print(f"sample_prob: {float((probs > 0).float().mean()):.3f} of nodes "
          f"reachable, mean prob {float(probs.mean()):.4f}")

    # 2) frequency partition with a per-partition hot cache
    with tempfile.TemporaryDirectory() as out:
        part = FrequencyPartitioner(
            out, num_parts=2, num_nodes=n, edge_index=edge_index,
            node_feat=feats, probs=[probs.cpu(), probs.cpu().flip(0)],
            cache_ratio=args.cache_ratio)
        part.partition()
        _, graph0, nfeat0, _, node_pb, _ = load_partition(out, 0)
        print(f"partition0: {graph0.edge_index.size(1)} edges, "
              f"{nfeat0.ids.numel()} feature rows, "
              f"cache {nfeat0.cache_ids.numel()} rows")

    # 3) plain training with hot rows first (sort_by_in_degree reorder)
    reordered, id2index = sort_by_in_degree(feats, 0.5, topo)
    ds = Dataset()
    ds.graph = g
    ds.node_features = Feature(reordered, split_ratio=0.5,
                               device=0 if has_gpu else None,
                               with_gpu=has_gpu, id2index=id2index)
    ds.node_labels = labels.to(device)
    loader = NeighborLoader(ds, fanout, input_nodes=torch.arange(n),
                            batch_size=args.batch_size, shuffle=True,
                            device=device, to_device=device)
    model = GraphSAGE(args.feat_dim, 128, len(fanout),
                      out_channels=10).to(device)
    opt = torch.optim.Adam(model.parameters(), lr=1e-3)
    t0 = time.time()
    it = iter(loader)
    for step in range(args.steps):
        try:
            data = next(it)
        except StopIteration:
            it = iter(loader)
            data = next(it)
        opt.zero_grad(set_to_none=True)
        out = model(data.x, data.edge_index, data.num_sampled_nodes,
                    data.num_sampled_edges)[:data.batch_size]
        loss = F.cross_entropy(out, data.y[:data.batch_size])
        loss.backward()
        opt.step()
    if has_gpu:
        torch.cuda.synchronize()
    print(f"trained {args.steps} steps, "
          f"{args.steps / (time.time() - t0):.1f} steps/s, "
          f"final loss {float(loss):.4f}")


if __name__ == "__main__":
    main()
