#!/usr/bin/env python3
"""Single-kernel microbench harness — the target for rocprofv3 --pmc runs.

rocprofv3 PMC collection serializes every dispatch, so counters must be
collected over a few dispatches of ONE kernel, not the full training
pipeline.  This harness builds flagship-shaped inputs
for one chosen op and runs exactly --iters dispatches of it.

    gpurun -- 'cd /tmp && export TMPDIR=/tmp && cd $GRAFT_REPO_ROOT && \
      rocprofv3 --pmc SQ_INSTS_MFMA,FETCH_SIZE,WRITE_SIZE --kernel-trace \
      --stats -d gpurun_out/pmc -- \
      python tools/kernel_microbench.py --op seg_mean_cat --iters 20'
"""
import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def build_flagship_shapes(op, device):
    """Layer-1-of-flagship shapes: ~590k rows, 890k edges, F=100/256."""
    torch.manual_seed(0)
    n_src, n_tgt, E, F = 590_000, 130_000, 890_000, 256
    if op in ("seg_mean", "seg_mean_cat"):
        x = torch.randn(n_src, F, device=device)
        tgt = torch.sort(torch.randint(0, n_tgt, (E,), device=device))[0]
        src = torch.randint(0, n_src, (E,), device=device)
        return (x, tgt, src, n_tgt)
    if op == "gat_fused":
        H, C = 4, 64
        h = torch.randn(n_src, H, C, device=device)
        att = torch.randn(H, C, device=device)
        tgt = torch.sort(torch.randint(0, n_tgt, (E,), device=device))[0]
        src = torch.randint(0, n_src, (E,), device=device)
        return (h, att, tgt, src, n_tgt)
    if op == "mfma_gemm":
        A = torch.randn(130_000, 512, device=device)
        B = torch.randn(128, 512, device=device)
        return (A, B)
    if op == "gemm_bt_bf16":
        A = torch.randn(164_000, 200, device=device).to(torch.bfloat16)
        W = torch.randn(256, 200, device=device).to(torch.bfloat16)
        b = torch.randn(256, device=device)
        return (A, W, b)
    if op == "gemm_kt_bf16":
        A = torch.randn(164_000, 256, device=device).to(torch.bfloat16)
        B = torch.randn(164_000, 200, device=device).to(torch.bfloat16)
        return (A, B)
    if op == "seg_mean_cat_bf16":
        n_src, n_tgt, E, F = 590_000, 130_000, 890_000, 100
        x = torch.randn(n_src, F, device=device).to(torch.bfloat16)
        tgt = torch.sort(torch.randint(0, n_tgt, (E,), device=device))[0]
        src = torch.randint(0, n_src, (E,), device=device)
        return (x, tgt, src, n_tgt)
    if op == "gather":
        feats = torch.randn(2_449_029, 100, device=device)
        rows = torch.randint(0, feats.size(0), (736_000,), device=device)
        return (feats, rows)
    if op == "sample":
        n, deg = 2_449_029, 25
        indptr = torch.arange(0, (n + 1) * deg, deg, device=device)
        indices = torch.randint(0, n, (n * deg,), device=device)
        seeds = torch.randint(0, n, (131_072,), device=device)
        return (indptr, indices, seeds)
    if op in ("sample_weighted", "sample_weighted_nr"):
        n, deg = 500_000, 50
        indptr = torch.arange(0, (n + 1) * deg, deg, device=device)
        indices = torch.randint(0, n, (n * deg,), device=device)
        w = torch.rand(n * deg, device=device)
        seeds = torch.randint(0, n, (131_072,), device=device)
        return (indptr, indices, w, seeds)
    if op == "sample_weighted_hub":
        # power-law-ish: 1000 hub rows of degree 10k + filler rows
        n_hub, hub_deg = 1000, 10_000
        n = n_hub
        indptr = torch.arange(0, (n + 1) * hub_deg, hub_deg, device=device)
        indices = torch.randint(0, 1 << 20, (n * hub_deg,), device=device)
        w = torch.rand(n * hub_deg, device=device)
        seeds = torch.randint(0, n, (8192,), device=device)
        return (indptr, indices, w, seeds)
    raise SystemExit(f"unknown op {op}")


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--op", required=True,
                    choices=["seg_mean", "seg_mean_cat", "gat_fused",
                             "mfma_gemm", "gather", "sample",
                             "sample_weighted", "sample_weighted_nr",
                             "sample_weighted_hub", "gemm_bt_bf16",
                             "gemm_kt_bf16", "seg_mean_cat_bf16"])
    ap.add_argument("--iters", type=int, default=20)
    ap.add_argument("--warmup", type=int, default=3)
    args = ap.parse_args()
    assert torch.cuda.is_available(), "microbench needs a GPU"
    device = torch.device("cuda", 0)
    import glt_amd
    from glt_amd import _C

    inp = build_flagship_shapes(args.op, device)

    op = args.op
    if args.op == "seg_mean":
        from glt_amd.ops.segment import _boundaries
        x, tgt, src, n_tgt = inp
        off = torch.searchsorted(tgt, _boundaries(n_tgt, device))
        fn = lambda: _C.segment_mean_fwd(x, src, off, n_tgt)
    elif args.op == "seg_mean_cat":
        from glt_amd.ops.segment import _boundaries
        x, tgt, src, n_tgt = inp
        off = torch.searchsorted(tgt, _boundaries(n_tgt, device))
        fn = lambda: _C.segment_mean_cat_fwd(x, src, off, n_tgt)
    elif args.op == "gat_fused":
        from glt_amd.ops.segment import _boundaries
        h, att, tgt, src, n_tgt = inp
        off = torch.searchsorted(tgt, _boundaries(n_tgt, device))
        fn = lambda: _C.gat_fused_fwd(h, h, att, att, src, off, 0.2)
    elif args.op == "mfma_gemm":
        A, B = inp
        fn = lambda: _C.sage_gemm(A, B, None, False)
    elif args.op == "gemm_bt_bf16":
        A, W, b = inp
        fn = lambda: _C.gemm_bt_bf16(A, W, b, True, False)
    elif args.op == "gemm_kt_bf16":
        A, B = inp
        fn = lambda: _C.gemm_kt_bf16(A, B, True)
    elif args.op == "seg_mean_cat_bf16":
        from glt_amd.ops.segment import _boundaries
        x, tgt, src, n_tgt = inp
        off = torch.searchsorted(tgt, _boundaries(n_tgt, device))
        fn = lambda: _C.segment_mean_cat_fwd(x, src, off, n_tgt)
    elif args.op == "gather":
        feats, rows = inp
        store = _C.UnifiedFeatureStore(0)
        store.append(feats)
        fn = lambda: store.gather(rows)
    elif op == "sample":
        indptr, indices, seeds = inp
        fn = lambda: _C.sample_neighbors(indptr, indices, seeds, 15)
    else:  # weighted variants
        indptr, indices, w, seeds = inp
        rep = op != "sample_weighted_nr"
        fn = lambda: _C.sample_neighbors(indptr, indices, seeds, 15,
                                         edge_weights=w, weighted=True,
                                         replace=rep)

    for _ in range(args.warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.iters):
        fn()
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / args.iters
    print(f"{args.op}: {dt * 1e6:.1f} us/iter over {args.iters} iters")


if __name__ == "__main__":
    main()
