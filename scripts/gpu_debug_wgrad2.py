#!/usr/bin/env python3
"""Debug: per-param defer-vs-immediate diffs across the full resnet18."""
import torch
from horizonml_amd import ops as _ops
from horizonml_amd.engine.flat import FlatParamManager
from horizonml_amd.models import resnet18
from horizonml_amd.models._functional_gpu import cross_entropy

C = _ops.extension()
dev = torch.device("cuda", 0)
grads = []
for defer in (False, True):
    torch.manual_seed(0)
    model = resnet18(num_classes=10).to(dev)
    mgr = FlatParamManager(model, dev)
    C.set_wgrad_defer(defer)
    torch.manual_seed(1)
    x = torch.randn(16, 3, 32, 32, device=dev).to(
        memory_format=torch.channels_last).to(torch.bfloat16)
    y = torch.randint(0, 10, (16,), device=dev)
    cross_entropy(model(x), y).backward()
    print("pending before flush:", C.wgrad_pending())
    C.flush_wgrad()
    torch.cuda.synchronize()
    grads.append({n: p.grad.clone() for n, p in model.named_parameters()})
    C.set_wgrad_defer(False)
for n in grads[0]:
    if "conv" not in n and "downsample" not in n:
        continue
    a, b = grads[0][n], grads[1][n]
    r = ((a - b).norm() / a.norm().clamp_min(1e-12)).item()
    flag = " <-- BAD" if r > 1e-3 else ""
    print(f"{n:<32} rel={r:.3e} |a|={a.norm().item():.3f}{flag}")
