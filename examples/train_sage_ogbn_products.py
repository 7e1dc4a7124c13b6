#!/usr/bin/env python3
"""Single-GPU supervised GraphSAGE on ogbn-products (capability parity:
reference examples/train_sage_ogbn_products.py).

With --synthetic (default here: the environment has no network for OGB
downloads) a random graph of the ogbn-products shape is used; pass
--dataset-root pointing at a prepared ogb dataset directory to train on the
real data (expects obg's processed tensors saved as .pt files).
"""
import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.nn.functional as F

import glt_amd
from glt_amd import Dataset, NeighborLoader
from glt_amd.models import GraphSAGE


def load_dataset(args, device):
    ds = Dataset()
    if args.dataset_root and os.path.isdir(args.dataset_root):
        edge_index = torch.load(os.path.join(args.dataset_root,
                                             "edge_index.pt"))
        feats = torch.load(os.path.join(args.dataset_root, "node_feat.pt"))
        labels = torch.load(os.path.join(args.dataset_root, "labels.pt"))
        split = torch.load(os.path.join(args.dataset_root, "split.pt"))
        train_idx = split["train"]
        n = feats.size(0)
    else:
        print("using synthetic ogbn-products-shaped data")
        n, e = 2_449_029, 61_859_140
        if not torch.cuda.is_available():
            n, e = 20_000, 400_000
        gen = device if torch.cuda.is_available() else torch.device("cpu")
        src = torch.randint(0, n, (e,), device=gen)
        dst = torch.randint(0, n, (e,), device=gen)
        edge_index = torch.stack([torch.cat([src, dst]),
                                  torch.cat([dst, src])]).cpu()
        feats = torch.randn(n, 100)
        labels = torch.randint(0, 47, (n,))
        train_idx = torch.randperm(n)[:196_615]
    mode = "CUDA" if torch.cuda.is_available() else "CPU"
    ds.init_graph(edge_index=edge_index, graph_mode=mode, num_nodes=n,
                  device=device.index)
    if args.dtype == "bf16":
        feats = feats.to(torch.bfloat16)
    ds.init_node_features(feats, split_ratio=args.split_ratio,
                          device=device.index,
                          with_gpu=torch.cuda.is_available())
    ds.init_node_labels(labels)
    return ds, train_idx


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--dataset-root", type=str, default="")
    ap.add_argument("--epochs", type=int, default=3)
    ap.add_argument("--batch-size", type=int, default=1024)
    ap.add_argument("--fanout", type=str, default="15,10,5")
    ap.add_argument("--hidden", type=int, default=256)
    ap.add_argument("--split-ratio", type=float, default=1.0)
    ap.add_argument("--dtype", type=str, default="bf16",
                    choices=["fp32", "bf16"],
                    help="bf16 (default): manual mixed precision — "
                         "features stored bf16, bf16 MFMA GEMMs/segment "
                         "kernels, fp32 master weights (accuracy parity: "
                         "profiles/r02_convergence.md)")
    ap.add_argument("--lr", type=float, default=0.003)
    args = ap.parse_args()

    device = torch.device("cuda", 0) if torch.cuda.is_available() \
        else torch.device("cpu")
    glt_amd.seed_everything(42)
    ds, train_idx = load_dataset(args, device)
    fanout = [int(x) for x in args.fanout.split(",")]
    loader = NeighborLoader(ds, fanout, input_nodes=train_idx,
                            batch_size=args.batch_size, shuffle=True,
                            device=device, to_device=device, prefetch=3)
    model = GraphSAGE(ds.node_features.size(1), args.hidden, len(fanout),
                      out_channels=47).to(device)
    opt = torch.optim.Adam(model.parameters(), lr=args.lr)

    for epoch in range(args.epochs):
        model.train()
        t0 = time.time()
        total_loss = total_correct = total = 0
        for data in loader:
            opt.zero_grad(set_to_none=True)
            out = model(data.x, data.edge_index, data.num_sampled_nodes,
                        data.num_sampled_edges)[:data.batch_size]
            y = data.y[:data.batch_size]
            loss = F.cross_entropy(out.float(), y)
            loss.backward()
            opt.step()
            total_loss += loss.item()
            total_correct += int((out.argmax(-1) == y).sum())
            total += y.numel()
        if torch.cuda.is_available():
            torch.cuda.synchronize()
        dt = time.time() - t0
        print(f"epoch {epoch}: loss {total_loss / max(total // args.batch_size, 1):.4f} "
              f"acc {total_correct / max(total, 1):.4f} "
              f"epoch_time {dt:.2f}s "
              f"batches/s {total / args.batch_size / dt:.1f}")


if __name__ == "__main__":
    main()
