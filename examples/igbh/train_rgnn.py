#!/usr/bin/env python3
"""Heterogeneous RGAT/RSAGE training on an IGBH-shaped synthetic graph
(capability parity: reference examples/igbh/dist_train_rgnn.py /
MLPerf IGBH harness, minus the real dataset which needs a network).

Node types: paper, author, institute, fos; edge types mirror IGBH:
  (paper, cites, paper), (author, writes, paper) reversed as
  (paper, rev_writes, author), (author, affiliated, institute),
  (paper, topic, fos).
Seeds are papers; 2-hop [10, 5] fan-out; checkpoint every --ckpt-steps via
glt_amd.utils.save_ckpt (the reference's MLPerf script does the same).
"""
import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.dirname(
    os.path.abspath(__file__)))))

import torch
import torch.nn.functional as F

import glt_amd
from glt_amd import Dataset, NeighborLoader
from glt_amd.models import RGNN
from glt_amd.utils import load_ckpt, save_ckpt


def build(args, device):
    n_paper, n_author, n_inst, n_fos = (args.papers, args.papers // 2,
                                        args.papers // 50,
                                        args.papers // 10)
    has_gpu = device.type == "cuda"

    def rnd_edges(ns, nd, e):
        return torch.stack([torch.randint(0, ns, (e,)),
                            torch.randint(0, nd, (e,))])

    e = args.papers * 10
    edges = {
        ("paper", "cites", "paper"): rnd_edges(n_paper, n_paper, e),
        ("paper", "rev_writes", "author"): rnd_edges(n_paper, n_author, e),
        ("author", "affiliated", "institute"): rnd_edges(n_author, n_inst,
                                                         e // 10),
        ("paper", "topic", "fos"): rnd_edges(n_paper, n_fos, e // 2),
    }
    ds = Dataset()
    ds.init_graph(edge_index=edges,
                  graph_mode="CUDA" if has_gpu else "CPU",
                  num_nodes={"paper": n_paper, "author": n_author,
                             "institute": n_inst, "fos": n_fos},
                  device=device.index)
    dim = args.feat_dim
    ds.init_node_features(
        {"paper": torch.randn(n_paper, dim),
         "author": torch.randn(n_author, dim),
         "institute": torch.randn(n_inst, dim),
         "fos": torch.randn(n_fos, dim)},
        split_ratio=1.0 if has_gpu else 0.0, device=device.index,
        with_gpu=has_gpu)
    ds.init_node_labels({"paper": torch.randint(0, args.classes,
                                                (n_paper,))})
    return ds


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--papers", type=int, default=100_000)
    ap.add_argument("--feat-dim", type=int, default=128)
    ap.add_argument("--classes", type=int, default=19)
    ap.add_argument("--hidden", type=int, default=256)
    ap.add_argument("--heads", type=int, default=4)
    ap.add_argument("--model", type=str, default="rgat",
                    choices=["rgat", "rsage"])
    ap.add_argument("--fanout", type=str, default="10,5")
    ap.add_argument("--batch-size", type=int, default=512)
    ap.add_argument("--epochs", type=int, default=2)
    ap.add_argument("--steps-per-epoch", type=int, default=50)
    ap.add_argument("--ckpt-dir", type=str, default="")
    ap.add_argument("--ckpt-steps", type=int, default=100)
    args = ap.parse_args()

    has_gpu = torch.cuda.is_available()
    if not has_gpu:
        args.papers = min(args.papers, 5_000)
    device = torch.device("cuda", 0) if has_gpu else torch.device("cpu")
    glt_amd.seed_everything(1)
    ds = build(args, device)
    fanout = [int(x) for x in args.fanout.split(",")]
    loader = NeighborLoader(
        ds, fanout, input_nodes=("paper", torch.arange(args.papers)),
        batch_size=args.batch_size, shuffle=True, device=device,
        to_device=device)
    model = RGNN(list(ds.graph.keys()), args.feat_dim, args.hidden,
                 args.classes, num_layers=len(fanout),
                 n_heads=args.heads, model=args.model).to(device)
    opt = torch.optim.Adam(model.parameters(), lr=1e-3)
    step = 0
    if args.ckpt_dir:
        load_ckpt(args.ckpt_dir, model, opt)
    for epoch in range(args.epochs):
        t0 = time.time()
        nb = 0
        for data in loader:
            opt.zero_grad(set_to_none=True)
            out = model(data.x_dict, data.edge_index_dict,
                        predict_type="paper")
            bs = data["paper"].batch_size
            loss = F.cross_entropy(out[:bs], data["paper"].y[:bs])
            loss.backward()
            opt.step()
            nb += 1
            step += 1
            if args.ckpt_dir and step % args.ckpt_steps == 0:
                save_ckpt(step, args.ckpt_dir, model, opt, epoch)
            if nb >= args.steps_per_epoch:
                break
        print(f"epoch {epoch}: loss {float(loss):.4f} "
              f"{nb / (time.time() - t0):.1f} batches/s")


if __name__ == "__main__":
    main()
