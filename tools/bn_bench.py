#!/usr/bin/env python
"""Microbench: FusedBNAct2d vs nn.BatchNorm2d(+relu)(+add) on ResNet-50
shapes (NHWC fp32). Prints per-op fwd / fwd+bwd times."""
import os
import sys
import time

import torch
import torch.nn as nn
import torch.nn.functional as F

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from dear_pytorch_amd.ops.fused_bn import FusedBNAct2d  # noqa: E402

SHAPES = [  # (N, C, H, W, res?)
    (64, 64, 112, 112, False),
    (64, 256, 56, 56, True),
    (64, 512, 28, 28, True),
    (64, 1024, 14, 14, True),
    (64, 2048, 7, 7, True),
]


def timeit(fn, iters=int(os.environ.get("BN_ITERS","20")), warmup=int(os.environ.get("BN_WARMUP","5")) ):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e3


def main():
    dev = torch.device("cuda:0")
    for N, C, H, W, res in SHAPES:
        x = torch.randn(N, C, H, W, device=dev) \
            .to(memory_format=torch.channels_last).requires_grad_(True)
        r = torch.randn_like(x) if res else None
        fused = FusedBNAct2d(C, relu=True).to(dev)
        plain = nn.BatchNorm2d(C).to(dev)
        g = torch.randn(N, C, H, W, device=dev) \
            .to(memory_format=torch.channels_last)

        def f_fwd():
            with torch.no_grad():
                fused(x, residual=r)

        def p_fwd():
            with torch.no_grad():
                y = plain(x)
                if r is not None:
                    y = y + r
                F.relu(y)

        def f_full():
            y = fused(x, residual=r)
            y.backward(g)
            x.grad = None

        def p_full():
            y = plain(x)
            if r is not None:
                y = y + r
            y = F.relu(y)
            y.backward(g)
            x.grad = None

        ff, pf = timeit(f_fwd), timeit(p_fwd)
        fb, pb = timeit(f_full), timeit(p_full)
        mb = N * C * H * W * 4 / 1e6
        print(f"N{N} C{C:5d} {H:3d}x{W:<3d} ({mb:6.0f} MB) "
              f"fwd fused {ff:7.3f} plain {pf:7.3f} | "
              f"fwd+bwd fused {fb:7.3f} plain {pb:7.3f} ms", flush=True)


if __name__ == "__main__":
    main()
