"""Quantify deterministic-mode cost: fast vs deterministic step time."""
import time
import torch
from horizonml_amd import ops as _ops
from horizonml_amd.engine.flat import FlatParamManager, HorizonAdam
from horizonml_amd.models import build_model
from horizonml_amd.models._functional_gpu import cross_entropy

C = _ops.extension()
dev = torch.device("cuda", 0)
for det in (False, True):
    C.set_deterministic(det)
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
    s = torch.cuda.Stream(); s.wait_stream(torch.cuda.current_stream())
    with torch.cuda.stream(s):
        for _ in range(3):
            l = step()
    torch.cuda.current_stream().wait_stream(s)
    torch.cuda.synchronize(); del l
    g = torch.cuda.CUDAGraph()
    with torch.cuda.graph(g):
        step()
    for _ in range(20):
        g.replay()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(200):
        g.replay()
    torch.cuda.synchronize()
    t1 = time.perf_counter()
    print(f"det={det}: {(t1-t0)/200*1e3:.3f} ms/step")
C.set_deterministic(False)
