"""Weight-gradient kernel isolation sweep (ResNet50@224 + CIFAR shapes):
achieved TF/s + effective GB/s per conv shape through wgrad_only.

Usage: python scripts/gpu_wgrad_sweep.py [batch]
"""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from horizonml_amd import ops

SHAPES = [
    # (in_ch, out_ch, k, stride, hw)
    (3, 64, 7, 2, 224),
    (64, 64, 3, 1, 56),
    (64, 256, 1, 1, 56),
    (256, 64, 1, 1, 56),
    (128, 128, 3, 1, 28),
    (512, 128, 1, 1, 28),
    (256, 256, 3, 1, 14),
    (1024, 256, 1, 1, 14),
    (512, 512, 3, 1, 7),
    (64, 64, 3, 1, 8),     # CIFAR layer1 (bs arg applies)
]


def main():
    C_ = ops.extension()
    bs = int(sys.argv[1]) if len(sys.argv) > 1 else 32
    print(f"batch={bs}")
    for cin, cout, k, s, hw in SHAPES:
        ho = (hw + 2 * (k // 2) - k) // s + 1
        M = bs * ho * ho
        x = torch.randn(bs, cin, hw, hw, device="cuda") \
            .to(memory_format=torch.channels_last).to(torch.bfloat16)
        dz = torch.randn(bs, cout, ho, ho, device="cuda") \
            .to(memory_format=torch.channels_last).to(torch.bfloat16)
        it = 30
        for _ in range(5):
            C_.wgrad_only(x, dz, cout, k, k, s, k // 2, False, None)
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(it):
            C_.wgrad_only(x, dz, cout, k, k, s, k // 2, False, None)
        torch.cuda.synchronize()
        dt = (time.perf_counter() - t0) / it
        flops = 2.0 * M * cout * cin * k * k
        # traffic: X gathered once per k3 tile row + Dz re-read per k3 tile
        tx = (cin * k * k + 63) // 64
        gb = (M * cin * k * k * 2 + M * cout * 2 * tx) / 1e9
        print(f"conv {cin:>4}x{hw:>3} -> {cout:>4} k{k}s{s} M={M:>7} "
              f"Kd={cin * k * k:>5}: {dt * 1e6:8.1f} us "
              f"{flops / dt / 1e12:7.1f} TF/s  {gb / dt:7.0f} GB/s(gross)")


if __name__ == "__main__":
    main()
