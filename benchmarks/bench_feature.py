#!/usr/bin/env python3
"""Feature-lookup micro-benchmark: gather GB/s from the UnifiedFeatureStore
(mirrors reference benchmarks/api/bench_feature.py, split_ratio knob)."""
import argparse
import json
import time

import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--rows", type=int, default=2_449_029)
    ap.add_argument("--dim", type=int, default=100)
    ap.add_argument("--batch", type=int, default=250_000)
    ap.add_argument("--split-ratio", type=float, default=1.0)
    ap.add_argument("--iters", type=int, default=30)
    ap.add_argument("--dtype", type=str, default="float32")
    args = ap.parse_args()

    import glt_amd
    from glt_amd.data import Feature

    assert torch.cuda.is_available(), "GPU micro-benchmark"
    dtype = getattr(torch, args.dtype)
    feats = torch.randn(args.rows, args.dim).to(dtype)
    f = Feature(feats, split_ratio=args.split_ratio, device=0, with_gpu=True)
    f.lazy_init()
    ids = torch.randint(0, args.rows, (args.batch,), device="cuda")
    for _ in range(3):
        out = f[ids]
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.iters):
        out = f[ids]
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    bytes_moved = args.iters * args.batch * args.dim * out.element_size() * 2
    print(json.dumps({
        "metric": "feature_gather_GBps",
        "value": round(bytes_moved / dt / 1e9, 2),
        "split_ratio": args.split_ratio,
        "rows_per_sec_M": round(args.iters * args.batch / dt / 1e6, 2),
        "dtype": args.dtype,
        "dim": args.dim,
    }))


if __name__ == "__main__":
    main()
