"""Training-step loop at an MFMA-meaningful shape for PMC capture.

The CIFAR parity workload (32x32, collapsing to 1x1 spatial) cannot be
MFMA-bound at any batch size — ResNet18/CIFAR needs ~43 GFLOP/step at
bs=1024, i.e. ~17 us of math against a 2.5 PFLOP/s bf16 peak — so MFMA%
there only measures launch/latency floor (profiles/README.md).  This script
runs the ImageNet-shaped ResNet50 configuration (BASELINE.json config #5's
model/shape) where the implicit-GEMM convs have real arithmetic intensity,
as the operating point for the north-star "MFMA shown with rocprof
counters" evidence.

Usage: python scripts/gpu_mfma_bench.py [bs] [image] [steps] [model]
"""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from horizonml_amd.engine.flat import FlatParamManager, HorizonAdam
from horizonml_amd.models import build_model
from horizonml_amd.models._functional_gpu import cross_entropy


def main():
    bs = int(sys.argv[1]) if len(sys.argv) > 1 else 32
    size = int(sys.argv[2]) if len(sys.argv) > 2 else 224
    steps = int(sys.argv[3]) if len(sys.argv) > 3 else 10
    name = sys.argv[4] if len(sys.argv) > 4 else "resnet50"
    dev = torch.device("cuda", 0)
    torch.manual_seed(0)
    model = build_model(name, num_classes=10).to(dev)
    mgr = FlatParamManager(model, dev)
    opt = HorizonAdam(mgr, lr=1e-3)
    x = torch.randn(bs, 3, size, size, device=dev) \
        .to(memory_format=torch.channels_last).to(torch.bfloat16)
    y = torch.randint(0, 10, (bs,), device=dev)

    def step():
        loss = cross_entropy(model(x), y)
        loss.backward()
        opt.step()
        return loss

    for _ in range(3):
        step()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(steps):
        step()
    torch.cuda.synchronize()
    t1 = time.perf_counter()
    print(f"{name} bs={bs} image={size}: "
          f"{(t1 - t0) / steps * 1000:.2f} ms/step, "
          f"{bs * steps / (t1 - t0):.1f} img/s")


if __name__ == "__main__":
    main()
