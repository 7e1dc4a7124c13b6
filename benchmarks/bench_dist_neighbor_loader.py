#!/usr/bin/env python3
"""Distributed neighbor-loader benchmark: 2 trainer ranks on localhost,
each with spawned sampling workers over RPC (mirrors reference
benchmarks/api/bench_dist_neighbor_loader.py on one node)."""
import argparse
import json
import multiprocessing as mp
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def run(rank, world, port, args, q):
    import glt_amd
    from glt_amd.data import Feature
    from glt_amd.distributed import (DistDataset, DistNeighborLoader,
                                     MpDistSamplingWorkerOptions,
                                     init_worker_group)
    from glt_amd.partition import GLTPartitionBook
    from glt_amd.utils import get_free_port

    glt_amd.seed_everything(rank)
    n, e = args.nodes, args.edges
    g = torch.Generator()
    g.manual_seed(7)
    src = torch.randint(0, n, (e,), generator=g)
    dst = torch.randint(0, n, (e,), generator=g)
    book = torch.randint(0, world, (n,), generator=g, dtype=torch.uint8)
    mask = book[src] == rank
    init_worker_group(world, rank)
    has_gpu = torch.cuda.is_available()
    device = torch.device("cuda", 0) if has_gpu else torch.device("cpu")
    ds = DistDataset(num_partitions=world, partition_idx=rank)
    ds.init_graph(edge_index=torch.stack([src[mask], dst[mask]]),
                  graph_mode="CUDA" if has_gpu else "CPU", num_nodes=n,
                  device=0 if has_gpu else None)
    ds.node_pb = GLTPartitionBook(book)
    feats = torch.randn(n, args.feat_dim)
    local = torch.nonzero(book.long() == rank).flatten()
    id2index = torch.full((n,), -1, dtype=torch.long)
    id2index[local] = torch.arange(local.numel())
    ds.node_features = Feature(feats[local], split_ratio=1.0 if has_gpu
                               else 0.0, device=0 if has_gpu else None,
                               with_gpu=has_gpu, id2index=id2index)
    ds._node_feat_pb = ds.node_pb
    opts = MpDistSamplingWorkerOptions(
        num_workers=args.sampling_workers, master_addr="127.0.0.1",
        master_port=port, channel_size="512MB", channel_capacity=64,
        pin_memory=has_gpu)
    fanout = [int(x) for x in args.fanout.split(",")]
    loader = DistNeighborLoader(ds, fanout, input_nodes=local,
                                batch_size=args.batch_size, shuffle=True,
                                to_device=device, worker_options=opts)
    # warmup epoch fraction
    t0 = time.perf_counter()
    nb = tot_edges = 0
    for data in loader:
        nb += 1
        tot_edges += data.edge_index.size(1)
        if nb >= args.batches:
            break
    dt = time.perf_counter() - t0
    loader.shutdown()
    q.put((rank, nb / dt, tot_edges / dt / 1e6))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--ranks", type=int, default=2)
    ap.add_argument("--nodes", type=int, default=1_000_000)
    ap.add_argument("--edges", type=int, default=20_000_000)
    ap.add_argument("--feat-dim", type=int, default=128)
    ap.add_argument("--fanout", type=str, default="15,10,5")
    ap.add_argument("--batch-size", type=int, default=1024)
    ap.add_argument("--batches", type=int, default=50)
    ap.add_argument("--sampling-workers", type=int, default=2)
    args = ap.parse_args()
    if not torch.cuda.is_available():
        args.nodes, args.edges, args.batches = 50_000, 500_000, 10
    from glt_amd.utils import get_free_port

    ctx = mp.get_context("spawn")
    port = get_free_port()
    q = ctx.Queue()
    ps = [ctx.Process(target=run, args=(r, args.ranks, port, args, q))
          for r in range(args.ranks)]
    for p in ps:
        p.start()
    res = [q.get(timeout=600) for _ in range(args.ranks)]
    for p in ps:
        p.join(timeout=30)
        if p.is_alive():
            p.terminate()
    print(json.dumps({
        "metric": "dist_loader_batches_per_sec (aggregate)",
        "value": round(sum(r[1] for r in res), 2),
        "sampled_edges_per_sec_M": round(sum(r[2] for r in res), 2),
        "ranks": args.ranks,
        "sampling_workers_per_rank": args.sampling_workers,
    }))


if __name__ == "__main__":
    main()
