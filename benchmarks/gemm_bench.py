#!/usr/bin/env python3
"""Microbenchmark: bflc_amd MFMA GEMM vs torch (hipBLASLt/rocBLAS).

Times the shapes the FL models actually run (FEMNIST / ResNet im2col
GEMMs) plus square reference shapes. Within-process interleaved A/B
(guide §5.4 rule 24): alternate ours/torch per round, report medians.
"""
import argparse
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def time_fn(fn, iters=20, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    ts = []
    for _ in range(iters):
        t0 = time.perf_counter()
        fn()
        torch.cuda.synchronize()
        ts.append(time.perf_counter() - t0)
    ts.sort()
    return ts[len(ts) // 2]


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--iters", type=int, default=20)
    args = ap.parse_args()
    from bflc_amd.ops import functional as fn
    hip = fn.hip_ops()
    dev = torch.device("cuda:0")

    shapes = [
        # (M, K, N) — model shapes
        (802816, 9, 32),     # femnist conv1 as GEMM (batch 1024)
        (200704, 288, 64),   # femnist conv2
        (1024, 3136, 128),   # femnist fc1
        (65536, 576, 64),    # resnet20 stage3 conv
        (12544, 2304, 512),  # resnet50 mid conv (batch 16)
        # square reference shapes
        (2048, 2048, 2048),
        (4096, 4096, 4096),
        (8192, 8192, 8192),
    ]
    print(f"{'M':>8} {'K':>6} {'N':>6} | {'ours ms':>9} {'ours TF':>8} | "
          f"{'torch ms':>9} {'torch TF':>8} | ratio")
    for M, K, N in shapes:
        x = torch.randn(M, K, device=dev, dtype=torch.bfloat16)
        w = torch.randn(K, N, device=dev, dtype=torch.bfloat16)
        b = torch.zeros(N, device=dev, dtype=torch.bfloat16)
        flops = 2.0 * M * K * N
        t_ours, t_torch = [], []
        for _ in range(3):  # interleaved A/B rounds
            t_ours.append(time_fn(lambda: hip.linear_fwd(x, w, b),
                                  args.iters // 3 + 1, 2))
            t_torch.append(time_fn(lambda: torch.matmul(x, w),
                                   args.iters // 3 + 1, 2))
        to, tt = min(t_ours), min(t_torch)
        print(f"{M:>8} {K:>6} {N:>6} | {to*1e3:>9.3f} {flops/to/1e12:>8.1f} |"
              f" {tt*1e3:>9.3f} {flops/tt/1e12:>8.1f} | "
              f"{tt/to:>5.2f}x")

    # Operand-layout ablation (gemm_raw): staging differs per layout —
    # K-contiguous operands use 16-B vector LDS writes, the others scatter.
    print("\nlayout ablation (ours only):")
    for M, K, N in [(4096, 4096, 4096), (12544, 2304, 512)]:
        flops = 2.0 * M * K * N
        for ta, tb in [(False, False), (False, True),
                       (True, False), (True, True)]:
            A = torch.randn(*((K, M) if ta else (M, K)), device=dev,
                            dtype=torch.bfloat16)
            B = torch.randn(*((N, K) if tb else (K, N)), device=dev,
                            dtype=torch.bfloat16)
            t = time_fn(lambda: hip.gemm_raw(A, B, ta, tb), args.iters, 3)
            print(f"  {M}x{K}x{N} ta={int(ta)} tb={int(tb)}: "
                  f"{t*1e3:8.3f} ms {flops/t/1e12:7.1f} TF")


if __name__ == "__main__":
    main()
