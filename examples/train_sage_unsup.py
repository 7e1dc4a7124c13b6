#!/usr/bin/env3 python
"""Unsupervised GraphSAGE link prediction with binary negative sampling
and random-walk positives (capability parity: reference
examples/train_sage_unsup.py + the RANDOM_WALK sampling type it stubs)."""
import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

import glt_amd
from glt_amd import Dataset, LinkNeighborLoader
from glt_amd.sampler import NegativeSampling, NeighborSampler
from glt_amd.models import GraphSAGE, unsupervised_link_pred_loss


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--nodes", type=int, default=100_000)
    ap.add_argument("--edges", type=int, default=1_000_000)
    ap.add_argument("--feat-dim", type=int, default=64)
    ap.add_argument("--hidden", type=int, default=128)
    ap.add_argument("--fanout", type=str, default="10,5")
    ap.add_argument("--batch-size", type=int, default=512)
    ap.add_argument("--epochs", type=int, default=2)
    ap.add_argument("--walk-len", type=int, default=0,
                    help=">0: derive positive pairs from random walks "
                         "instead of raw edges")
    args = ap.parse_args()

    has_gpu = torch.cuda.is_available()
    if not has_gpu:
        args.nodes, args.edges = 5_000, 50_000
    device = torch.device("cuda", 0) if has_gpu else torch.device("cpu")
    glt_amd.seed_everything(0)

    n = args.nodes
    src = torch.randint(0, n, (args.edges,))
    dst = torch.randint(0, n, (args.edges,))
    ds = Dataset()
    ds.init_graph(edge_index=torch.stack([src, dst]),
                  graph_mode="CUDA" if has_gpu else "CPU", num_nodes=n,
                  device=0 if has_gpu else None)
    ds.init_node_features(torch.randn(n, args.feat_dim), split_ratio=1.0,
                          device=0 if has_gpu else None, with_gpu=has_gpu)

    fanout = [int(x) for x in args.fanout.split(",")]
    if args.walk_len > 0:
        # random-walk positives: pair each walk start with its walk nodes
        sampler = NeighborSampler(ds.get_graph(), fanout, device=device)
        starts = torch.randint(0, n, (args.edges // 10,), device=device)
        walks = sampler.random_walk(starts, args.walk_len)
        pos_src = walks[:, :-1].reshape(-1)
        pos_dst = walks[:, 1:].reshape(-1)
        eli = torch.stack([pos_src, pos_dst]).cpu()
    else:
        eli = torch.stack([src, dst])

    loader = LinkNeighborLoader(
        ds, fanout, edge_label_index=eli,
        neg_sampling=NegativeSampling("binary"),
        batch_size=args.batch_size, shuffle=True, device=device,
        to_device=device, prefetch=3 if torch.cuda.is_available() else 0)
    model = GraphSAGE(args.feat_dim, args.hidden, len(fanout)).to(device)
    opt = torch.optim.Adam(model.parameters(), lr=1e-3)

    for epoch in range(args.epochs):
        t0 = time.time()
        total = nb = 0
        for data in loader:
            opt.zero_grad(set_to_none=True)
            h = model(data.x, data.edge_index)
            loss = unsupervised_link_pred_loss(h, data.edge_label_index,
                                               data.edge_label)
            loss.backward()
            opt.step()
            total += float(loss)
            nb += 1
            if nb >= 50:
                break
        print(f"epoch {epoch}: loss {total / max(nb, 1):.4f} "
              f"({nb / (time.time() - t0):.1f} batches/s)")


if __name__ == "__main__":
    main()
