"""KRSC->RSCK batched permute isolation bench (flat-engine RSCK refresh).

Usage: HZ_PERMUTE_V2={0,1} python scripts/gpu_permute_bench.py
"""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from horizonml_amd.engine.flat import FlatParamManager
from horizonml_amd.models import resnet18, resnet50


def main():
    print(f"HZ_PERMUTE_V2={os.environ.get('HZ_PERMUTE_V2', '(default 1)')}")
    for name, fn in [("resnet18", resnet18), ("resnet50", resnet50)]:
        model = fn(num_classes=10).cuda()
        mgr = FlatParamManager(model, torch.device("cuda"))
        for _ in range(30):
            mgr.refresh_rsck()
        torch.cuda.synchronize()
        it = 300
        t0 = time.perf_counter()
        for _ in range(it):
            mgr.refresh_rsck()
        torch.cuda.synchronize()
        dt = (time.perf_counter() - t0) / it
        gb = 2 * mgr.rsck.numel() * 2 / 1e9  # read + write bf16
        print(f"{name}: {dt * 1e6:8.1f} us  {gb / dt:7.1f} GB/s "
              f"({mgr.rsck.numel() / 1e6:.1f}M weights)")


if __name__ == "__main__":
    main()
