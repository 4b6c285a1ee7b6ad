import torch
from horizonml_amd.models import build_model
from horizonml_amd.models.resnet import Bottleneck

torch.manual_seed(0)
cpu = build_model("resnet50", num_classes=10)
gpu = build_model("resnet50", num_classes=10)
gpu.load_state_dict(cpu.state_dict())
gpu = gpu.cuda()
x = torch.randn(4, 3, 64, 64)
xg = x.cuda().to(memory_format=torch.channels_last).to(torch.bfloat16)
ref = cpu(x)
out_fused = gpu(xg)
# disable fusion: monkeypatch _fusable
orig = Bottleneck._fusable
Bottleneck._fusable = lambda self: False
out_unfused = gpu(xg)
Bottleneck._fusable = orig
def rel(a, b):
    a = a.float().cpu(); b = b.float().cpu()
    return ((a - b).norm() / b.norm().clamp_min(1e-12)).item()
print("fused   vs cpu:", rel(out_fused, ref))
print("unfused vs cpu:", rel(out_unfused, ref))
print("fused vs unfused:", rel(out_fused, out_unfused))
print("ref norm", ref.norm().item(), "out norm", out_fused.float().norm().item())
