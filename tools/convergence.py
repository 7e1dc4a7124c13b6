#!/usr/bin/env python3
"""Convergence evidence at the ogbn-products scale/config.

The reference's single hard published number is 0.7870 test accuracy for
GraphSAGE on ogbn-products ([15,10,5], batch 1024, 20 epochs —
reference examples/train_sage_ogbn_products.py:16).  This environment
has no network, so this harness builds a LABEL-CORRELATED synthetic
graph of the exact products shape (2.45M nodes, 124M directed edges, 47
classes, 100-dim features, power-law degrees) and runs the same
training protocol with train/val/test splits sized like products
(196,615 / 39,323 / 2,213,091), reporting the per-epoch accuracy curve
for fp32 and bf16 so reduced precision shows its accuracy parity.

Construction: communities = the 47 classes; each edge endpoint pair is
drawn intra-community with probability --homophily (products-like
assortativity), else uniformly; per-node lognormal degree weights give
the hub-heavy histogram; features are a noisy class embedding
(100-dim random class prototype + N(0, --noise) per node).

Run (GPU): python tools/convergence.py --epochs 20 --dtype bf16
"""
import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.nn.functional as F


def build(args, device):
    import glt_amd
    from glt_amd.data import Graph, Topology

    glt_amd.seed_everything(args.seed)
    g = torch.Generator(device=device)
    g.manual_seed(args.seed)
    n, e, k = args.nodes, args.edges, args.classes
    comm = torch.randint(0, k, (n,), device=device, generator=g)
    # degree weights (lognormal) within the community-sorted order
    w = (torch.randn(n, device=device, generator=g) * args.skew).exp()
    # order nodes by community so intra-community draws are range lookups
    order = torch.argsort(comm)
    comm_sorted = comm[order]
    counts = torch.bincount(comm, minlength=k)
    starts = torch.cat([torch.zeros(1, dtype=torch.long, device=device),
                        torch.cumsum(counts, 0)[:-1]])
    w_sorted = w[order]
    cdf_all = torch.cumsum(w_sorted, 0)
    tot_all = cdf_all[-1].clone()

    src_list, dst_list = [], []
    chunk = 20_000_000
    for s in range(0, e, chunk):
        m = min(chunk, e - s)
        u = torch.rand(m, device=device, generator=g)
        src = torch.searchsorted(cdf_all, u * tot_all).clamp_(0, n - 1)
        # destination: intra-community (weighted within the community
        # range) with prob homophily, else global weighted draw
        intra = torch.rand(m, device=device, generator=g) < args.homophily
        c = comm_sorted[src]
        lo = starts[c]
        hi = lo + counts[c]
        base = torch.where(lo > 0, cdf_all[(lo - 1).clamp(min=0)],
                           torch.zeros_like(u))
        base = torch.where(lo > 0, base, torch.zeros_like(base))
        span = cdf_all[hi - 1] - base
        ud = torch.rand(m, device=device, generator=g)
        dst_in = torch.searchsorted(cdf_all, base + ud * span)
        dst_in = dst_in.clamp_(0, n - 1)
        dst_gl = torch.searchsorted(
            cdf_all, torch.rand(m, device=device, generator=g) * tot_all
        ).clamp_(0, n - 1)
        dst = torch.where(intra, dst_in, dst_gl)
        src_list.append(src)
        dst_list.append(dst)
    src = torch.cat(src_list)
    dst = torch.cat(dst_list)
    del src_list, dst_list, cdf_all
    row = torch.cat([src, dst])
    col = torch.cat([dst, src])
    del src, dst
    perm = torch.argsort(row * n + col)
    row_s, col_s = row[perm], col[perm]
    del row, col, perm
    counts_r = torch.bincount(row_s, minlength=n)
    indptr = torch.zeros(n + 1, dtype=torch.long, device=device)
    torch.cumsum(counts_r, 0, out=indptr[1:])
    del row_s
    topo = Topology((indptr, col_s), input_layout="CSR", layout="CSR",
                    auto_edge_ids=False)
    graph = Graph(topo, mode="CUDA", device=device.index)
    graph._indptr, graph._indices = topo.indptr, topo.indices
    graph._edge_ids = graph._edge_weights = None
    graph._lazy_done = True

    # features: noisy class prototypes (sorted space), labels = community
    proto = torch.randn(k, args.feat_dim, device=device, generator=g)
    feats = proto[comm_sorted] + args.noise * torch.randn(
        n, args.feat_dim, device=device, generator=g)
    labels = comm_sorted
    deg_stats = {
        "mean_deg": float(counts_r.float().mean()),
        "median_deg": float(counts_r.float().median()),
        "max_deg": int(counts_r.max()),
    }
    return graph, feats, labels, deg_stats


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--nodes", type=int, default=2_449_029)
    ap.add_argument("--edges", type=int, default=61_859_140)
    ap.add_argument("--feat-dim", type=int, default=100)
    ap.add_argument("--classes", type=int, default=47)
    ap.add_argument("--hidden", type=int, default=256)
    ap.add_argument("--fanout", type=str, default="15,10,5")
    ap.add_argument("--batch-size", type=int, default=1024)
    ap.add_argument("--epochs", type=int, default=20)
    ap.add_argument("--dtype", type=str, default="bf16",
                    choices=["fp32", "bf16"])
    ap.add_argument("--homophily", type=float, default=0.6)
    ap.add_argument("--skew", type=float, default=1.2)
    ap.add_argument("--noise", type=float, default=2.0)
    ap.add_argument("--seed", type=int, default=7)
    ap.add_argument("--train", type=int, default=196_615)
    ap.add_argument("--val", type=int, default=39_323)
    ap.add_argument("--test-eval", type=int, default=200_000,
                    help="test nodes evaluated (sampled from the test "
                         "split for eval speed)")
    args = ap.parse_args()
    assert torch.cuda.is_available(), "convergence run needs the GPU"
    device = torch.device("cuda", 0)
    torch.cuda.set_device(device)

    import glt_amd
    from glt_amd import Dataset, NeighborLoader
    from glt_amd.data import Feature
    from glt_amd.models import GraphSAGE

    t0 = time.perf_counter()
    graph, feats, labels, deg_stats = build(args, device)
    fdt = torch.bfloat16 if args.dtype == "bf16" else torch.float32
    ds = Dataset()
    ds.graph = graph
    ds.node_features = Feature(feats.to(fdt).cpu(), split_ratio=1.0,
                               device=0, with_gpu=True)
    del feats
    ds.node_labels = labels
    n = args.nodes
    perm = torch.randperm(n, device=device)
    train_idx = perm[: args.train]
    val_idx = perm[args.train: args.train + args.val]
    test_idx = perm[args.train + args.val:][: args.test_eval]
    build_s = time.perf_counter() - t0

    fanout = [int(x) for x in args.fanout.split(",")]
    model = GraphSAGE(args.feat_dim, args.hidden, len(fanout),
                      out_channels=args.classes).to(device)
    opt = torch.optim.Adam(model.parameters(), lr=3e-3, fused=True)
    loader = NeighborLoader(ds, fanout, input_nodes=train_idx,
                            batch_size=args.batch_size, shuffle=True,
                            device=device, to_device=device, prefetch=3)

    @torch.no_grad()
    def evaluate(idx):
        model.eval()
        ev = NeighborLoader(ds, fanout, input_nodes=idx,
                            batch_size=4096, device=device,
                            to_device=device, prefetch=3)
        correct = total = 0
        for data in ev:
            out = model(data.x, data.edge_index, data.num_sampled_nodes,
                        data.num_sampled_edges)[:data.batch_size]
            y = data.y[:data.batch_size]
            correct += int((out.argmax(-1) == y).sum())
            total += y.numel()
        model.train()
        return correct / max(total, 1)

    curve = []
    for epoch in range(args.epochs):
        te = time.perf_counter()
        for data in loader:
            opt.zero_grad(set_to_none=True)
            out = model(data.x, data.edge_index, data.num_sampled_nodes,
                        data.num_sampled_edges)[:data.batch_size]
            loss = F.cross_entropy(out.float(), data.y[:data.batch_size])
            loss.backward()
            opt.step()
        torch.cuda.synchronize()
        ep_s = time.perf_counter() - te
        acc = evaluate(val_idx)
        curve.append({"epoch": epoch + 1, "val_acc": round(acc, 4),
                      "epoch_s": round(ep_s, 3)})
        print(json.dumps(curve[-1]), flush=True)
    test_acc = evaluate(test_idx)
    print(json.dumps({
        "metric": "GraphSAGE synthetic-products test accuracy",
        "dtype": args.dtype, "epochs": args.epochs,
        "test_acc": round(test_acc, 4),
        "final_val_acc": curve[-1]["val_acc"],
        "epoch_s_mean": round(sum(c["epoch_s"] for c in curve)
                              / len(curve), 3),
        "build_s": round(build_s, 1),
        "deg_stats": deg_stats,
        "config": {"nodes": args.nodes, "edges": args.edges,
                   "homophily": args.homophily, "skew": args.skew,
                   "noise": args.noise, "fanout": fanout,
                   "batch": args.batch_size,
                   "train/val/test": [args.train, args.val,
                                      args.test_eval]},
    }))


if __name__ == "__main__":
    main()
