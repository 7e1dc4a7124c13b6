#!/usr/bin/env python3
"""Distributed supervised GraphSAGE over a partitioned dataset (capability
parity: reference examples/distributed/dist_train_sage_supervised.py).

Run on every node:
  python dist_train_sage_supervised.py --num-nodes 2 --node-rank R \\
      --master-addr A --dataset-root /path/to/partitions

Each trainer rank loads its partition, spawns sampling workers streaming
batches through a pinned shm channel, and trains under DDP (RCCL)."""
import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.dirname(
    os.path.abspath(__file__)))))

import torch
import torch.distributed as dist
import torch.nn.functional as F


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--dataset-root", type=str, required=True)
    ap.add_argument("--num-nodes", type=int, default=1)
    ap.add_argument("--node-rank", type=int, default=0)
    ap.add_argument("--master-addr", type=str, default="127.0.0.1")
    ap.add_argument("--master-port", type=int, default=29400)
    ap.add_argument("--rpc-port", type=int, default=29401)
    ap.add_argument("--epochs", type=int, default=3)
    ap.add_argument("--batch-size", type=int, default=1024)
    ap.add_argument("--fanout", type=str, default="15,10,5")
    ap.add_argument("--hidden", type=int, default=256)
    ap.add_argument("--classes", type=int, default=47)
    ap.add_argument("--num-sampling-workers", type=int, default=2)
    ap.add_argument("--train-idx-file", type=str, default="")
    ap.add_argument("--label-file", type=str, default="")
    args = ap.parse_args()

    import glt_amd
    from glt_amd.distributed import (DistDataset, DistNeighborLoader,
                                     MpDistSamplingWorkerOptions,
                                     init_worker_group)
    from glt_amd.models import GraphSAGE

    rank, world = args.node_rank, args.num_nodes
    device = torch.device("cuda", rank % max(torch.cuda.device_count(), 1)) \
        if torch.cuda.is_available() else torch.device("cpu")
    if device.type == "cuda":
        torch.cuda.set_device(device)

    # training process group (DDP gradient all-reduce over RCCL)
    os.environ["MASTER_ADDR"] = args.master_addr
    os.environ["MASTER_PORT"] = str(args.master_port)
    dist.init_process_group("nccl" if device.type == "cuda" else "gloo",
                            rank=rank, world_size=world)
    init_worker_group(world, rank)

    ds = DistDataset(edge_dir="out")
    ds.load(args.dataset_root, rank,
            graph_mode="CUDA" if device.type == "cuda" else "CPU",
            whole_node_label_file=args.label_file or None,
            device=device.index)
    train_idx = (torch.load(args.train_idx_file)
                 if args.train_idx_file else
                 torch.arange(0, 100_000))

    opts = MpDistSamplingWorkerOptions(
        num_workers=args.num_sampling_workers,
        master_addr=args.master_addr, master_port=args.rpc_port,
        channel_size="1GB", channel_capacity=128, pin_memory=True)
    fanout = [int(x) for x in args.fanout.split(",")]
    loader = DistNeighborLoader(ds, fanout, input_nodes=train_idx,
                                batch_size=args.batch_size, shuffle=True,
                                to_device=device, worker_options=opts)

    in_dim = ds.node_features.size(1) if ds.node_features is not None else 0
    model = GraphSAGE(in_dim, args.hidden, len(fanout),
                      out_channels=args.classes).to(device)
    model = torch.nn.parallel.DistributedDataParallel(
        model, device_ids=[device.index] if device.type == "cuda" else None)
    opt = torch.optim.Adam(model.parameters(), lr=1e-3)

    for epoch in range(args.epochs):
        t0 = time.time()
        nb = 0
        for data in loader:
            opt.zero_grad(set_to_none=True)
            out = model(data.x, data.edge_index, data.num_sampled_nodes,
                        data.num_sampled_edges)[:data.batch_size]
            loss = F.cross_entropy(out, data.y[:data.batch_size].long())
            loss.backward()
            opt.step()
            nb += 1
        if device.type == "cuda":
            torch.cuda.synchronize()
        dist.barrier()
        if rank == 0:
            print(f"epoch {epoch}: {nb} local batches, "
                  f"{time.time() - t0:.2f}s")
    loader.shutdown()
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
