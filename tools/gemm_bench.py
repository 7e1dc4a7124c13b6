#!/usr/bin/env python3
"""bf16 MFMA GEMM vs hipBLASLt at the flagship projection shapes.

Times (and numerics-checks) the three hot GEMM shapes of the GraphSAGE
step in bf16 on both paths:
  fwd   C[M,N] = A[M,K] @ W[N,K]^T (+bias, relu)
  dW    [N,K'] = dy[M,N]^T @ x[M,K']  (+ db)   -- split-K kernel
  dgrad dx[M,K] = dy[M,N] @ W[N,K]

Run: gpurun -- 'python tools/gemm_bench.py'
"""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def t(fn, iters=50, warmup=10):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6


def main():
    assert torch.cuda.is_available()
    import glt_amd  # noqa: F401
    from glt_amd import _C

    dev = torch.device("cuda", 0)
    torch.manual_seed(0)
    shapes = [
        ("L0 fwd", 164_000, 200, 256),
        ("L1 fwd", 16_300, 512, 256),
        ("L2 fwd", 1_024, 512, 47),
    ]
    print("== fwd: C = A @ W^T + bias (relu) ==")
    for name, M, K, N in shapes:
        A = torch.randn(M, K, device=dev).to(torch.bfloat16)
        W = (torch.randn(N, K, device=dev) / K ** 0.5).to(torch.bfloat16)
        b = torch.randn(N, device=dev)
        C1 = _C.gemm_bt_bf16(A, W, b, True, False)
        C2 = torch.relu(torch.nn.functional.linear(
            A.float(), W.float(), b))
        err = (C1.float() - C2).abs().max().item()
        scale = C2.abs().max().item()
        us1 = t(lambda: _C.gemm_bt_bf16(A, W, b, True, False))
        us2 = t(lambda: torch.relu_(torch.nn.functional.linear(
            A, W, b.to(torch.bfloat16))))
        fl = 2 * M * K * N
        print(f"{name} M={M} K={K} N={N}: mfma {us1:7.1f}us "
              f"({fl / us1 / 1e6:6.1f} TF/s)  blaslt {us2:7.1f}us "
              f"({fl / us2 / 1e6:6.1f} TF/s)  maxerr {err:.3f} "
              f"(|C|max {scale:.1f})")

    print("== dW: A^T @ B (+db) ==")
    for name, Kb, M2, N2 in [("L0 dW", 164_000, 256, 200),
                             ("L1 dW", 16_300, 256, 512),
                             ("L2 dW", 1_024, 47, 512)]:
        A = torch.randn(Kb, M2, device=dev).to(torch.bfloat16)
        B = torch.randn(Kb, N2, device=dev).to(torch.bfloat16)
        C1, db1 = _C.gemm_kt_bf16(A, B, True)
        C2 = A.float().t() @ B.float()
        db2 = A.float().sum(0)
        err = (C1 - C2).abs().max().item() / max(C2.abs().max().item(), 1)
        dberr = (db1 - db2).abs().max().item() / max(
            db2.abs().max().item(), 1)
        us1 = t(lambda: _C.gemm_kt_bf16(A, B, True))
        us2 = t(lambda: (A.t() @ B).float())
        us3 = t(lambda: A.float().sum(0))
        fl = 2 * Kb * M2 * N2
        print(f"{name} Kb={Kb} M={M2} N={N2}: mfma+db {us1:7.1f}us "
              f"({fl / us1 / 1e6:6.1f} TF/s)  blaslt {us2:7.1f}us "
              f"+colsum {us3:5.1f}us  relerr {err:.4f} dbrelerr "
              f"{dberr:.4f}")

    print("== dgrad: dy @ W ==")
    for name, M, N, K in [("L1 dgrad", 16_300, 256, 512)]:
        dy = torch.randn(M, N, device=dev).to(torch.bfloat16)
        W = (torch.randn(N, K, device=dev) / K ** 0.5).to(torch.bfloat16)
        Wt = W.t().contiguous()
        C1 = _C.gemm_bt_bf16(dy, Wt, None, False, False)
        C2 = dy.float() @ W.float()
        err = (C1.float() - C2).abs().max().item()
        us1 = t(lambda: _C.gemm_bt_bf16(dy, W.t().contiguous(), None,
                                        False, False))
        us2 = t(lambda: dy @ W)
        fl = 2 * M * K * N
        print(f"{name}: mfma {us1:7.1f}us ({fl / us1 / 1e6:6.1f} TF/s)  "
              f"blaslt {us2:7.1f}us ({fl / us2 / 1e6:6.1f} TF/s)  "
              f"maxerr {err:.3f}")


if __name__ == "__main__":
    main()
