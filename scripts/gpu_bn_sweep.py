"""BN-backward reduce kernel isolation sweep: scalar (HZ_BN_V8=0) vs
vectorized (HZ_BN_V8=1) per (M, C) shape, GB/s achieved.

Usage: HZ_BN_V8={0,1} python scripts/gpu_bn_sweep.py
"""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from horizonml_amd import ops

SHAPES = [
    # r50@224 bs=32 conv-output (M, C) pairs
    (32 * 112 * 112, 64),
    (32 * 56 * 56, 64),
    (32 * 56 * 56, 256),
    (32 * 28 * 28, 128),
    (32 * 28 * 28, 512),
    (32 * 14 * 14, 256),
    (32 * 14 * 14, 1024),
    (32 * 7 * 7, 512),
    (32 * 7 * 7, 2048),
    # CIFAR bs=64
    (64 * 8 * 8, 64),
    (64 * 1 * 1, 512),
    # slab-rule shapes: mid/small M with big C (r50@224 layer3/4 leftovers,
    # mobilenet head)
    (32 * 14 * 14, 512),
    (32 * 7 * 7, 1024),
    (64 * 4 * 4, 256),
    (64 * 2 * 2, 512),
    (49 * 8, 1280),
]


def main():
    C_ = ops.extension()
    print(f"HZ_BN_V8={os.environ.get('HZ_BN_V8', '(default 1)')} mask=2")
    for M, C in SHAPES:
        dy = torch.randn(M, C, device="cuda").bfloat16()
        y = torch.randn(M, C, device="cuda").bfloat16()
        x = torch.randn(M, C, device="cuda").bfloat16()
        mean = torch.randn(C, device="cuda")
        invstd = torch.rand(C, device="cuda") + 0.5
        gamma = torch.randn(C, device="cuda")
        beta = torch.randn(C, device="cuda")
        sdz = torch.zeros(C, device="cuda")
        sdzx = torch.zeros(C, device="cuda")
        it = 50
        for _ in range(5):
            C_.bn_reduce_bench(dy, y, x, mean, invstd, gamma, beta, sdz,
                               sdzx, M, C, 2)
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(it):
            C_.bn_reduce_bench(dy, y, x, mean, invstd, gamma, beta, sdz,
                               sdzx, M, C, 2)
        torch.cuda.synchronize()
        dt = (time.perf_counter() - t0) / it
        # correctness vs fp32 torch reference (mask 2: bn(x) > 0 gate)
        sdz.zero_(); sdzx.zero_()
        C_.bn_reduce_bench(dy, y, x, mean, invstd, gamma, beta, sdz,
                           sdzx, M, C, 2)
        g = dy.float() * ((gamma * invstd * x.float()
                           + (beta - mean * gamma * invstd)) > 0)
        r1 = g.sum(0)
        r2 = (g * (x.float() - mean) * invstd).sum(0)
        e1 = (sdz - r1).abs().max() / (r1.abs().max() + 1e-6)
        e2 = (sdzx - r2).abs().max() / (r2.abs().max() + 1e-6)
        ok = "ok" if max(e1.item(), e2.item()) < 2e-2 else "MISMATCH"
        gb = 2 * M * C * 2 / 1e9  # dy + x read, bf16
        print(f"M={M:>9} C={C:>5}: {dt * 1e6:8.1f} us  {gb / dt:7.1f} GB/s  {ok}")


if __name__ == "__main__":
    main()
