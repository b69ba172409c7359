#!/usr/bin/env python3
"""Microbenchmark of the fused BN kernels across the IMHN's layer shapes.

    IBP_BN_V8=1 python scripts/bn_perf.py   # vectorized path
    IBP_BN_V8=0 python scripts/bn_perf.py   # scalar path

Prints per-shape µs and effective bandwidth for bn_stats / bn_act_bwd /
bn_act_bwd_apply / bn_act_fwd.
"""
from __future__ import annotations

import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from improved_body_parts_amd.ops._backend import hip_extension  # noqa: E402

# (N, C, H, W) of the 4-stage IMHN @512^2 batch 16 (fwd conv outputs)
SHAPES = [
    (16, 64, 256, 256),   # stem
    (16, 128, 128, 128),  # backbone
    (16, 256, 128, 128),  # hourglass scale 0
    (16, 384, 64, 64),
    (16, 512, 32, 32),
    (16, 640, 16, 16),
    (16, 768, 8, 8),
]


def timeit(fn, iters=50, warmup=10):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6  # us


def main():
    assert torch.cuda.is_available()
    ext = hip_extension()
    v8 = os.environ.get("IBP_BN_V8", "1")
    print(f"IBP_BN_V8={v8}")
    print(f"{'shape':>22} {'stats':>9} {'bwd':>9} {'apply':>9} {'fwd':>9}  (us)")
    for (n, c, h, w) in SHAPES:
        x = torch.randn(n, c, h, w, device="cuda").bfloat16() \
            .contiguous(memory_format=torch.channels_last)
        dy = torch.randn_like(x)
        y = torch.randn_like(x)
        mean = torch.randn(c, device="cuda")
        invstd = torch.rand(c, device="cuda") + 0.5
        gamma = torch.randn(c, device="cuda")
        scale = torch.randn(c, device="cuda")
        shift = torch.randn(c, device="cuda")
        M = n * h * w

        t_stats = timeit(lambda: ext.bn_stats(x, c))
        dpre, sd, sx = ext.bn_act_bwd(dy, y, x, mean, invstd, 0.01, True, True, c)
        t_bwd = timeit(lambda: ext.bn_act_bwd(dy, y, x, mean, invstd, 0.01,
                                              True, True, c))
        t_apply = timeit(lambda: ext.bn_act_bwd_apply(dpre, x, mean, invstd,
                                                      gamma, sd, sx, c))
        t_fwd = timeit(lambda: ext.bn_act_fwd(x, scale, shift, None, 0.01, True))
        bytes_stats = M * c * 2
        bw = bytes_stats / (t_stats * 1e-6) / 1e12
        print(f"{str((n,c,h,w)):>22} {t_stats:9.2f} {t_bwd:9.2f} {t_apply:9.2f} "
              f"{t_fwd:9.2f}   stats_bw={bw:.2f} TB/s")


if __name__ == "__main__":
    main()
