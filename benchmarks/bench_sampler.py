#!/usr/bin/env python3
"""Neighbor-sampling micro-benchmark: sampled edges/sec (M).

Mirrors the reference harness metric (reference
benchmarks/api/bench_sampler.py:46-53) on a synthetic ogbn-products-shaped
graph: 3-hop [15,10,5], batch 1024, reporting sampled edges per second.
Modes: CUDA (HBM CSR) and ZERO_COPY (pinned-host UVA CSR).
"""
import argparse
import json
import time

import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--nodes", type=int, default=2_449_029)
    ap.add_argument("--edges", type=int, default=61_859_140)
    ap.add_argument("--batch-size", type=int, default=1024)
    ap.add_argument("--fanout", type=str, default="15,10,5")
    ap.add_argument("--iters", type=int, default=50)
    ap.add_argument("--mode", type=str, default="CUDA",
                    choices=["CUDA", "ZERO_COPY", "CPU"])
    ap.add_argument("--with-features", action="store_true")
    ap.add_argument("--feat-dim", type=int, default=100)
    args = ap.parse_args()

    import glt_amd
    from glt_amd.data import Feature, Graph, Topology
    from glt_amd.sampler import NeighborSampler, NodeSamplerInput

    glt_amd.seed_everything(0)
    has_gpu = torch.cuda.is_available()
    if not has_gpu:
        args.nodes, args.edges, args.iters = 20_000, 400_000, 5
        args.mode = "CPU"
    dev = torch.device("cuda", 0) if has_gpu and args.mode != "CPU" \
        else torch.device("cpu")
    gen = torch.device("cuda", 0) if has_gpu else torch.device("cpu")

    n, e = args.nodes, args.edges
    src = torch.randint(0, n, (e,), device=gen)
    dst = torch.randint(0, n, (e,), device=gen)
    row = torch.cat([src, dst])
    col = torch.cat([dst, src])
    perm = torch.argsort(row)
    row_s, col_s = row[perm], col[perm]
    indptr = torch.zeros(n + 1, dtype=torch.long, device=gen)
    torch.cumsum(torch.bincount(row_s, minlength=n), 0, out=indptr[1:])

    topo = Topology.__new__(Topology)
    topo.layout = "CSR"
    topo.edge_ids = None
    topo.edge_weights = None
    if args.mode == "CUDA":
        topo.indptr, topo.indices = indptr, col_s
    else:
        topo.indptr, topo.indices = indptr.cpu(), col_s.cpu()
    graph = Graph(topo, mode=args.mode, device=0 if has_gpu else None)
    if args.mode == "CUDA":
        graph._indptr, graph._indices = indptr, col_s
        graph._edge_ids = graph._edge_weights = None
        graph._lazy_done = True

    fanout = [int(x) for x in args.fanout.split(",")]
    sampler = NeighborSampler(graph, fanout, device=dev)
    feature = None
    if args.with_features:
        feats = torch.randn(n, args.feat_dim)
        feature = Feature(feats, split_ratio=1.0 if args.mode == "CUDA"
                          else 0.0, device=0, with_gpu=has_gpu)

    total_edges = 0
    total_nodes = 0
    # warmup
    for _ in range(3):
        seeds = torch.randint(0, n, (args.batch_size,), device=dev)
        out = sampler.sample_from_nodes(NodeSamplerInput(seeds))
    if has_gpu:
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.iters):
        seeds = torch.randint(0, n, (args.batch_size,), device=dev)
        out = sampler.sample_from_nodes(NodeSamplerInput(seeds))
        total_edges += out.row.numel()
        total_nodes += out.node.numel()
        if feature is not None:
            x = feature[out.node]
    if has_gpu:
        torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    print(json.dumps({
        "metric": "sampled_edges_per_sec_M",
        "value": round(total_edges / dt / 1e6, 2),
        "mode": args.mode,
        "with_features": args.with_features,
        "batches_per_sec": round(args.iters / dt, 2),
        "avg_edges_per_batch": total_edges // args.iters,
        "avg_nodes_per_batch": total_nodes // args.iters,
        "ms_per_batch": round(dt / args.iters * 1e3, 3),
    }))


if __name__ == "__main__":
    main()
