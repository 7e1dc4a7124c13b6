#!/usr/bin/env python3
"""Multi-GPU DP GraphSAGE via mp.spawn + DDP over RCCL (capability parity:
reference examples/multi_gpu/train_sage_ogbn_papers100m.py).

One process per GPU; every rank holds the full (synthetic) graph and its
HBM-resident feature shard; DDP all-reduce rides RCCL over xGMI."""
import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.dirname(
    os.path.abspath(__file__)))))

import torch
import torch.distributed as dist
import torch.multiprocessing as mp
import torch.nn.functional as F


def run(rank, world, args, port):
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", str(port))
    device = torch.device("cuda", rank)
    torch.cuda.set_device(device)
    dist.init_process_group("nccl", rank=rank, world_size=world)

    import glt_amd
    from glt_amd import Dataset, NeighborLoader
    from glt_amd.models import GraphSAGE

    glt_amd.seed_everything(42 + rank)
    n, e = args.nodes, args.edges
    g = torch.Generator(device=device)
    g.manual_seed(7)
    src = torch.randint(0, n, (e,), device=device, generator=g)
    dst = torch.randint(0, n, (e,), device=device, generator=g)
    ds = Dataset()
    ds.init_graph(edge_index=torch.stack([torch.cat([src, dst]),
                                          torch.cat([dst, src])]).cpu(),
                  graph_mode="CUDA", num_nodes=n, device=rank)
    ds.init_node_features(torch.randn(n, args.feat_dim), split_ratio=1.0,
                          device=rank)
    ds.init_node_labels(torch.randint(0, args.classes, (n,)))

    fanout = [int(x) for x in args.fanout.split(",")]
    seeds = torch.arange(rank, n, world)  # shard seeds across ranks
    loader = NeighborLoader(ds, fanout, input_nodes=seeds,
                            batch_size=args.batch_size, shuffle=True,
                            device=device, to_device=device, prefetch=3)
    model = GraphSAGE(args.feat_dim, args.hidden, len(fanout),
                      out_channels=args.classes).to(device)
    model = torch.nn.parallel.DistributedDataParallel(model,
                                                      device_ids=[rank])
    opt = torch.optim.Adam(model.parameters(), lr=1e-3)

    for epoch in range(args.epochs):
        t0 = time.time()
        nb = 0
        for data in loader:
            opt.zero_grad(set_to_none=True)
            out = model(data.x, data.edge_index, data.num_sampled_nodes,
                        data.num_sampled_edges)[:data.batch_size]
            loss = F.cross_entropy(out, data.y[:data.batch_size])
            loss.backward()
            opt.step()
            nb += 1
        torch.cuda.synchronize()
        dist.barrier()
        if rank == 0:
            dt = time.time() - t0
            print(f"epoch {epoch}: {nb * world} batches "
                  f"{nb * world / dt:.1f} batches/s  {dt:.2f}s")
    dist.destroy_process_group()


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int,
                    default=torch.cuda.device_count() or 1)
    ap.add_argument("--nodes", type=int, default=2_449_029)
    ap.add_argument("--edges", type=int, default=61_859_140)
    ap.add_argument("--feat-dim", type=int, default=100)
    ap.add_argument("--classes", type=int, default=47)
    ap.add_argument("--hidden", type=int, default=256)
    ap.add_argument("--fanout", type=str, default="15,10,5")
    ap.add_argument("--batch-size", type=int, default=1024)
    ap.add_argument("--epochs", type=int, default=2)
    args = ap.parse_args()
    from glt_amd.utils import get_free_port

    port = get_free_port()
    mp.spawn(run, args=(args.gpus, args, port), nprocs=args.gpus, join=True)


if __name__ == "__main__":
    main()
