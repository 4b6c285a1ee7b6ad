#!/usr/bin/env python3
"""Debug: stem-conv wgrad — deferred/batched vs immediate vs torch ref."""
import torch
import torch.nn.functional as F

from horizonml_amd import ops as _ops
from horizonml_amd.models.layers import ConvBNAct

C = _ops.extension()
dev = torch.device("cuda", 0)


def run(defer, cin=3, cout=64, k=7, s=2, hw=32, bs=16):
    torch.manual_seed(0)
    mod = ConvBNAct(cin, cout, k, stride=s, padding=k // 2).to(dev)
    # direct-grad mode: managed + pre-assigned grads
    mod._managed = True
    mod.weight_bf16 = mod.weight.data.to(torch.bfloat16)
    for p in mod.parameters():
        p.grad = torch.zeros_like(p, dtype=torch.float32)
    C.set_wgrad_defer(defer)
    torch.manual_seed(1)
    x = torch.randn(bs, cin, hw, hw)
    xg = x.cuda().to(memory_format=torch.channels_last).to(torch.bfloat16)
    y = mod(xg)
    y.float().square().mean().backward()
    C.flush_wgrad()
    torch.cuda.synchronize()
    C.set_wgrad_defer(False)
    return mod, x


# layer1-shaped conv first: defer (ms=8) vs immediate (ms=56), vec path
mi, _ = run(False, cin=64, cout=64, k=3, s=1, hw=8, bs=64)
md, _ = run(True, cin=64, cout=64, k=3, s=1, hw=8, bs=64)
g1, g2 = mi.weight.grad.float(), md.weight.grad.float()
print("[layer1] immediate vs defer rel:",
      ((g1 - g2).norm() / g1.norm()).item())

torch.manual_seed(123)
mod_i, x = run(False)
mod_d, _ = run(True)
gi = mod_i.weight.grad.float()
gd = mod_d.weight.grad.float()
print("immediate vs defer rel:",
      ((gi - gd).norm() / gi.norm()).item(),
      "max abs diff:", (gi - gd).abs().max().item())

# torch fp32 reference of the same conv+bn+relu backward
torch.manual_seed(0)
ref = torch.nn.Sequential(
    torch.nn.Conv2d(3, 64, 7, 2, 3, bias=False),
    torch.nn.BatchNorm2d(64), torch.nn.ReLU())
with torch.no_grad():
    # ConvBNAct init order: weight, bn_weight, bn_bias — replicate by seed
    torch.manual_seed(0)
    m2 = ConvBNAct(3, 64, 7, stride=2, padding=3)
    ref[0].weight.copy_(m2.weight.view(64, 7, 7, 3).permute(0, 3, 1, 2))
    ref[1].weight.copy_(m2.bn_weight)
    ref[1].bias.copy_(m2.bn_bias)
torch.manual_seed(1)
xr = x.clone().requires_grad_(False)
yr = ref(xr)
yr.square().mean().backward()
gr = ref[0].weight.grad.permute(0, 2, 3, 1).contiguous().view(64, 7, 7, 3)
print("immediate vs torch rel:", ((gi - gr.cuda()).norm() / gr.norm().cuda()).item())
print("defer    vs torch rel:", ((gd - gr.cuda()).norm() / gr.norm().cuda()).item())
