#!/usr/bin/env python3
"""Unsupervised link-prediction benchmark (BASELINE config #5: GraphSAGE
link-pred with binary negative sampling + random-walk positives)."""
import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--nodes", type=int, default=2_449_029)
    ap.add_argument("--edges", type=int, default=61_859_140)
    ap.add_argument("--feat-dim", type=int, default=100)
    ap.add_argument("--dtype", type=str, default="bf16",
                    choices=["fp32", "bf16"])
    ap.add_argument("--hidden", type=int, default=256)
    ap.add_argument("--fanout", type=str, default="10,5")
    ap.add_argument("--batch-size", type=int, default=1024)
    ap.add_argument("--steps", type=int, default=60)
    ap.add_argument("--warmup", type=int, default=10)
    ap.add_argument("--walk-len", type=int, default=0)
    args = ap.parse_args()

    import glt_amd
    from glt_amd import Dataset, LinkNeighborLoader
    from glt_amd.models import GraphSAGE, unsupervised_link_pred_loss
    from glt_amd.sampler import NegativeSampling, NeighborSampler

    has_gpu = torch.cuda.is_available()
    if not has_gpu:
        args.nodes, args.edges, args.steps, args.warmup = 20_000, 200_000, \
            5, 2
    device = torch.device("cuda", 0) if has_gpu else torch.device("cpu")
    gen = device if has_gpu else torch.device("cpu")
    glt_amd.seed_everything(0)
    n = args.nodes
    src = torch.randint(0, n, (args.edges,), device=gen)
    dst = torch.randint(0, n, (args.edges,), device=gen)
    ds = Dataset()
    ds.init_graph(edge_index=torch.stack([src, dst]).cpu(),
                  graph_mode="CUDA" if has_gpu else "CPU", num_nodes=n,
                  device=0 if has_gpu else None)
    fdt = torch.bfloat16 if args.dtype == "bf16" else torch.float32
    ds.init_node_features(torch.randn(n, args.feat_dim).to(fdt),
                          split_ratio=1.0 if has_gpu else 0.0,
                          device=0 if has_gpu else None, with_gpu=has_gpu)

    fanout = [int(x) for x in args.fanout.split(",")]
    if args.walk_len > 0:
        sampler = NeighborSampler(ds.get_graph(), fanout)
        starts = torch.randint(0, n, (args.edges // 20,), device=device)
        walks = sampler.random_walk(starts, args.walk_len)
        eli = torch.stack([walks[:, :-1].reshape(-1),
                           walks[:, 1:].reshape(-1)]).cpu()
    else:
        eli = torch.stack([src, dst]).cpu()
    loader = LinkNeighborLoader(
        ds, fanout, edge_label_index=eli,
        neg_sampling=NegativeSampling("binary"),
        batch_size=args.batch_size, shuffle=True, device=device,
        to_device=device, prefetch=3 if torch.cuda.is_available() else 0)
    model = GraphSAGE(args.feat_dim, args.hidden, len(fanout)).to(device)
    opt = torch.optim.Adam(model.parameters(), lr=1e-3)
    it = iter(loader)

    def step():
        nonlocal it
        try:
            data = next(it)
        except StopIteration:
            it = iter(loader)
            data = next(it)
        opt.zero_grad(set_to_none=True)
        h = model(data.x, data.edge_index)
        loss = unsupervised_link_pred_loss(h, data.edge_label_index,
                                           data.edge_label)
        loss.backward()
        opt.step()

    for _ in range(args.warmup):
        step()
    if has_gpu:
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        step()
    if has_gpu:
        torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    loader.shutdown()
    print(json.dumps({
        "metric": "unsup GraphSAGE link-pred batches/sec",
        "value": round(args.steps / dt, 3),
        "ms_per_step": round(dt / args.steps * 1e3, 3),
        "config": {"nodes": n, "fanout": fanout,
                   "batch_pos_edges": args.batch_size,
                   "neg": "binary x1", "walk_len": args.walk_len},
    }))


if __name__ == "__main__":
    main()
