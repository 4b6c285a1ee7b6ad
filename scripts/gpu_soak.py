"""Memory-stability soak: 3000 graph-replayed steps, assert no growth."""
import torch
from horizonml_amd.engine.flat import FlatParamManager, HorizonAdam
from horizonml_amd.models import build_model
from horizonml_amd.models._functional_gpu import cross_entropy

dev = torch.device("cuda", 0)
torch.manual_seed(0)
model = build_model("resnet18", num_classes=10).to(dev)
mgr = FlatParamManager(model, dev)
opt = HorizonAdam(mgr, lr=1e-3)
x = torch.randn(64, 3, 32, 32, device=dev).to(
    memory_format=torch.channels_last).to(torch.bfloat16)
y = torch.randint(0, 10, (64,), device=dev)

def step():
    loss = cross_entropy(model(x), y)
    loss.backward()
    opt.step()
    return loss

side = torch.cuda.Stream()
side.wait_stream(torch.cuda.current_stream())
with torch.cuda.stream(side):
    for _ in range(3):
        loss = step()
torch.cuda.current_stream().wait_stream(side)
torch.cuda.synchronize()
del loss
g = torch.cuda.CUDAGraph()
with torch.cuda.graph(g):
    ls = step()
g.replay()
torch.cuda.synchronize()
m0 = torch.cuda.memory_allocated()
for i in range(3000):
    g.replay()
torch.cuda.synchronize()
m1 = torch.cuda.memory_allocated()
print(f"3000 steps: mem {m0/1e6:.1f} -> {m1/1e6:.1f} MB, "
      f"loss {float(ls):.4f}")
assert m1 <= m0 + 1_000_000, "memory grew during replay soak"
assert torch.isfinite(ls), "non-finite loss after soak"
print("SOAK OK")
