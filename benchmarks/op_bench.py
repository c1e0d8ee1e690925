#!/usr/bin/env python3
"""Per-op timing sweep over the FL models' exact hot shapes.

Times each framework op (conv fwd/bwd, BN fwd/bwd, pool, the GEMM
shapes) with CUDA events and prints one JSON blob — a finer-grained
regression harness than rocprof kernel stats (which mix call sites):
run it before and after a kernel change and diff the per-shape numbers.

    python benchmarks/op_bench.py [--iters 50] [--model all]
"""
import argparse
import json
import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from bflc_amd.ops import functional as O


def timeit(fn, iters, warmup=10):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    s = torch.cuda.Event(enable_timing=True)
    e = torch.cuda.Event(enable_timing=True)
    s.record()
    for _ in range(iters):
        fn()
    e.record()
    torch.cuda.synchronize()
    return s.elapsed_time(e) * 1e3 / iters  # us


# (name, N, H, C, Kout, R, stride, pad) — the conv layers of the three
# image models at their bench batch sizes
CONVS = [
    ("fe.conv1", 1024, 28, 1, 32, 3, 1, 1),
    ("fe.conv2", 1024, 14, 32, 64, 3, 1, 1),
    ("r20.stem", 512, 32, 3, 16, 3, 1, 1),
    ("r20.b1", 512, 32, 16, 16, 3, 1, 1),
    ("r20.b2dn", 512, 32, 16, 32, 3, 2, 1),
    ("r20.b2", 512, 16, 32, 32, 3, 1, 1),
    ("r20.b3dn", 512, 16, 32, 64, 3, 2, 1),
    ("r20.b3", 512, 8, 64, 64, 3, 1, 1),
    ("r50.stem", 64, 224, 3, 64, 7, 2, 3),
    ("r50.c2.1x1", 64, 56, 64, 64, 1, 1, 0),
    ("r50.c2.3x3", 64, 56, 64, 64, 3, 1, 1),
    ("r50.c2.out", 64, 56, 64, 256, 1, 1, 0),
    ("r50.c3.3x3", 64, 28, 128, 128, 3, 1, 1),
    ("r50.c4.3x3", 64, 14, 256, 256, 3, 1, 1),
    ("r50.c5.3x3", 64, 7, 512, 512, 3, 1, 1),
]

# (name, M, K, N) GEMMs (linear layers + eval shapes)
GEMMS = [
    ("fe.fc1", 1024, 3136, 128),
    ("fe.fc2", 1024, 128, 62),
    ("r50.fc", 64, 2048, 1000),
    ("sq.2048", 2048, 2048, 2048),
]


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--iters", type=int, default=50)
    args = ap.parse_args()
    assert torch.cuda.is_available(), "op_bench needs a GPU"
    dev = torch.device("cuda", 0)
    torch.manual_seed(0)
    out = {}

    for name, n, h, c, k, r, stride, pad in CONVS:
        x = torch.randn(n, h, h, c, device=dev, dtype=torch.bfloat16)
        w = torch.randn(k, r, r, c, device=dev, dtype=torch.bfloat16) * 0.05
        b = torch.randn(k, device=dev, dtype=torch.bfloat16)
        h_ops = O.hip_ops()
        y, col, _, _ = h_ops.conv2d_fwd_bn(x, w, stride, pad)
        dy = torch.randn_like(y)
        col_arg = col if col.numel() else None
        out[f"conv.{name}.fwd"] = timeit(
            lambda: h_ops.conv2d_fwd_col(x, w, b, stride, pad, False,
                                         True), args.iters)
        out[f"conv.{name}.fwd_eval"] = timeit(
            lambda: h_ops.conv2d_fwd(x, w, b, stride, pad), args.iters)
        out[f"conv.{name}.bwd"] = timeit(
            lambda: h_ops.conv2d_bwd(x, w, dy, stride, pad, col_arg),
            args.iters)
        if k % 8 == 0:
            g = torch.rand(k, device=dev, dtype=torch.bfloat16) + 0.5
            be = torch.randn(k, device=dev, dtype=torch.bfloat16)
            yb, mean, invstd = h_ops.batchnorm_fwd(y, g, be, 1e-5, True,
                                                   None)
            out[f"bn.{name}.fwd"] = timeit(
                lambda: h_ops.batchnorm_fwd(y, g, be, 1e-5, True, None),
                args.iters)
            out[f"bn.{name}.bwd"] = timeit(
                lambda: h_ops.batchnorm_bwd(y, dy, mean, invstd, g, yb),
                args.iters)

    for name, m, k, n in GEMMS:
        x = torch.randn(m, k, device=dev, dtype=torch.bfloat16)
        w = torch.randn(k, n, device=dev, dtype=torch.bfloat16)
        b = torch.randn(n, device=dev, dtype=torch.bfloat16)
        dy = torch.randn(m, n, device=dev, dtype=torch.bfloat16)
        h_ops = O.hip_ops()
        out[f"gemm.{name}.fwd"] = timeit(
            lambda: h_ops.linear_fwd(x, w, b, False), args.iters)
        out[f"gemm.{name}.bwd"] = timeit(
            lambda: h_ops.linear_bwd(x, w, dy), args.iters)

    print(json.dumps({"unit": "us_per_call", "ops": out}, indent=1))


if __name__ == "__main__":
    main()
