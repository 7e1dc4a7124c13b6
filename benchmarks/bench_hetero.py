#!/usr/bin/env python3
"""Hetero RGAT/RSAGE end-to-end benchmark (BASELINE config #4: R-GAT on an
IGBH-shaped hetero graph).  JSON one-liner like bench.py."""
import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.nn.functional as F


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--papers", type=int, default=1_000_000)
    ap.add_argument("--feat-dim", type=int, default=128)
    ap.add_argument("--classes", type=int, default=19)
    ap.add_argument("--hidden", type=int, default=256)
    ap.add_argument("--heads", type=int, default=4)
    ap.add_argument("--model", type=str, default="rgat",
                    choices=["rgat", "rsage"])
    ap.add_argument("--fanout", type=str, default="10,5")
    ap.add_argument("--batch-size", type=int, default=512)
    ap.add_argument("--dtype", type=str, default="fp32",
                    choices=["fp32", "bf16"],
                    help="bf16: features stored bf16, bf16 GAT/segment "
                         "kernels + cast-linear over fp32 master params")
    ap.add_argument("--steps", type=int, default=60)
    ap.add_argument("--warmup", type=int, default=10)
    ap.add_argument("--prefetch", type=int, default=3)
    # pipeline diagnosis: time only the producer (loader) or only the
    # consumer (model step on one cached batch)
    ap.add_argument("--sampler-only", action="store_true")
    ap.add_argument("--model-only", action="store_true")
    args = ap.parse_args()

    import glt_amd
    from glt_amd import Dataset, NeighborLoader
    from glt_amd.models import RGNN

    has_gpu = torch.cuda.is_available()
    if not has_gpu:
        args.papers, args.steps, args.warmup = 10_000, 5, 2
    device = torch.device("cuda", 0) if has_gpu else torch.device("cpu")
    glt_amd.seed_everything(1)

    n_paper = args.papers
    n_author, n_inst, n_fos = n_paper // 2, n_paper // 50, n_paper // 10
    e = n_paper * 10

    def rnd(ns, nd, m):
        return torch.stack([torch.randint(0, ns, (m,)),
                            torch.randint(0, nd, (m,))])

    edges = {
        ("paper", "cites", "paper"): rnd(n_paper, n_paper, e),
        ("paper", "rev_writes", "author"): rnd(n_paper, n_author, e),
        ("author", "affiliated", "institute"): rnd(n_author, n_inst, e // 10),
        ("paper", "topic", "fos"): rnd(n_paper, n_fos, e // 2),
    }
    ds = Dataset()
    ds.init_graph(edge_index=edges,
                  graph_mode="CUDA" if has_gpu else "CPU",
                  num_nodes={"paper": n_paper, "author": n_author,
                             "institute": n_inst, "fos": n_fos},
                  device=0 if has_gpu else None)
    dim = args.feat_dim
    fdt = torch.bfloat16 if args.dtype == "bf16" else torch.float32
    ds.init_node_features(
        {"paper": torch.randn(n_paper, dim).to(fdt),
         "author": torch.randn(n_author, dim).to(fdt),
         "institute": torch.randn(n_inst, dim).to(fdt),
         "fos": torch.randn(n_fos, dim).to(fdt)},
        split_ratio=1.0 if has_gpu else 0.0,
        device=0 if has_gpu else None, with_gpu=has_gpu)
    ds.init_node_labels({"paper": torch.randint(0, args.classes,
                                                (n_paper,)).to(device)})

    fanout = [int(x) for x in args.fanout.split(",")]
    loader = NeighborLoader(
        ds, fanout, input_nodes=("paper", torch.arange(n_paper)),
        batch_size=args.batch_size, shuffle=True, device=device,
        to_device=device, prefetch=args.prefetch if has_gpu else 0)
    model = RGNN(list(ds.graph.keys()), dim, args.hidden, args.classes,
                 num_layers=len(fanout), n_heads=args.heads,
                 model=args.model).to(device)
    try:
        opt = torch.optim.Adam(model.parameters(), lr=1e-3, fused=has_gpu)
    except (RuntimeError, ValueError):
        opt = torch.optim.Adam(model.parameters(), lr=1e-3)
    it = iter(loader)

    cached = [next(it)] if args.model_only else None

    def step():
        nonlocal it
        if cached is not None:
            data = cached[0]
        else:
            try:
                data = next(it)
            except StopIteration:
                it = iter(loader)
                data = next(it)
        if args.sampler_only:
            return
        opt.zero_grad(set_to_none=True)
        out = model(data.x_dict, data.edge_index_dict,
                    predict_type="paper")
        bs = data["paper"].batch_size
        loss = F.cross_entropy(out[:bs].float(), data["paper"].y[:bs])
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
        "metric": f"{args.model} IGBH-shaped hetero train batches/sec",
        "value": round(args.steps / dt, 3),
        "ms_per_step": round(dt / args.steps * 1e3, 3),
        "config": {"dtype": args.dtype, "papers": n_paper, "feat_dim": dim,
                   "fanout": fanout, "batch": args.batch_size,
                   "model": args.model, "heads": args.heads},
    }))


if __name__ == "__main__":
    main()
