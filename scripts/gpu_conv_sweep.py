"""Per-shape conv kernel throughput sweep (fwd + dgrad), ResNet50@224
shapes.  Prints achieved TFLOP/s per shape so tile-selection thresholds
(HZ_TILE_FILL) can be tuned by measurement; run with HZ_TILE_FILL=999999
to force the 64x64 latency tile for an A/B.

Usage: python scripts/gpu_conv_sweep.py [batch]
"""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from horizonml_amd.models.layers import ConvBNAct

SHAPES = [
    # (in_ch, out_ch, k, stride, hw)   — ResNet50 @224 representative
    (3, 64, 7, 2, 224),
    (64, 64, 3, 1, 56),
    (64, 256, 1, 1, 56),
    (256, 64, 1, 1, 56),
    (128, 128, 3, 1, 28),
    (512, 128, 1, 1, 28),
    (256, 256, 3, 1, 14),
    (1024, 256, 1, 1, 14),
    (512, 512, 3, 1, 7),
    (2048, 512, 1, 1, 7),
]


def bench_one(cin, cout, k, s, hw, bs, iters=20):
    torch.manual_seed(0)
    mod = ConvBNAct(cin, cout, k, stride=s, act=True).cuda()
    x = torch.randn(bs, cin, hw, hw, device="cuda") \
        .to(memory_format=torch.channels_last).to(torch.bfloat16) \
        .requires_grad_(True)
    y = mod(x)
    gy = torch.randn_like(y)
    ho = y.shape[2]
    flops_fwd = 2.0 * bs * ho * ho * cout * cin * k * k

    def run_fwd():
        return mod(x)

    def run_bwd():
        yy = mod(x)
        yy.backward(gy)

    for fn, name, flops in [(run_fwd, "fwd", flops_fwd),
                            (run_bwd, "fwd+bwd", 3 * flops_fwd)]:
        for _ in range(3):
            fn()
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(iters):
            fn()
        torch.cuda.synchronize()
        dt = (time.perf_counter() - t0) / iters
        print(f"  {name:8s} {dt * 1e6:9.1f} us  {flops / dt / 1e12:8.1f} TF/s")


def main():
    bs = int(sys.argv[1]) if len(sys.argv) > 1 else 32
    fill = os.environ.get("HZ_TILE_FILL", "(default 192)")
    print(f"batch={bs} HZ_TILE_FILL={fill}")
    for cfg in SHAPES:
        cin, cout, k, s, hw = cfg
        m = bs * (hw // s) * (hw // s)
        print(f"conv {cin}x{hw}x{hw} -> {cout} k{k}s{s}  (M={m})")
        bench_one(*cfg, bs)


if __name__ == "__main__":
    main()
