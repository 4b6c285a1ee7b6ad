"""Per-dispatch floor: 100 dependent tiny kernels in one hipGraph."""
import time
import torch
from horizonml_amd import ops as _ops
C = _ops.extension()
dev = torch.device("cuda", 0)
t = torch.zeros(1, device=dev)
master = torch.zeros(4096, device=dev); grad = torch.zeros_like(master)
m = torch.zeros_like(master); v = torch.zeros_like(master)

def burst(n):
    for _ in range(n):
        C.adam_step(master, grad, m, v, None, t, 1e-3, 0.9, 0.999, 1e-8,
                    0.0, False, None, None, 1.0,
                    None, None, None)  # k_inc_step + k_adam (4096)

for _ in range(3):
    burst(50)
torch.cuda.synchronize()
g = torch.cuda.CUDAGraph()
with torch.cuda.graph(g):
    burst(50)   # 100 kernels total
g.replay(); torch.cuda.synchronize()
t0 = time.perf_counter()
for _ in range(50):
    g.replay()
torch.cuda.synchronize()
t1 = time.perf_counter()
per_kernel = (t1 - t0) / 50 / 100 * 1e6
print(f"100-kernel graph replay: {(t1-t0)/50*1e3:.3f} ms -> {per_kernel:.2f} us/dispatch")

# bigger grid tiny kernel: bn_apply eval on small tensor
x = torch.zeros(64, 64, 8, 8, device=dev, dtype=torch.bfloat16).to(memory_format=torch.channels_last)
