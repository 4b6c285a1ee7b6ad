"""Diagnostic: dump MFMA fragment-layout behavior if the GEMM test fails.

Runs A=I (asymmetric B) and small structured GEMMs, printing actual-vs-ref
blocks so a fragment transpose is identifiable from one gpurun log.
"""
import torch

from horizonml_amd import ops

_C = ops.extension()

torch.manual_seed(0)
# A = identity: C should equal B^T's first rows = B columns... C = I @ B^T = B^T
A = torch.eye(32, 64)
B = torch.arange(32 * 64).float().reshape(32, 64) / 100.0  # asymmetric
C = _C.gemm_bf16(A.cuda().bfloat16(), B.cuda().bfloat16()).float().cpu()
ref = A @ B.T
print("identity-A test: rel err",
      ((C - ref).norm() / ref.norm()).item())
print("C[0:4,0:4]:\n", C[:4, :4])
print("ref[0:4,0:4]:\n", ref[:4, :4])
print("C.T[0:4,0:4]:\n", C.T[:4, :4])

# delta test: A one-hot at (2, 5) -> C[2, j] = B[j, 5]
A2 = torch.zeros(16, 32)
A2[2, 5] = 1.0
B2 = torch.randn(16, 32)
C2 = _C.gemm_bf16(A2.cuda().bfloat16(), B2.cuda().bfloat16()).float().cpu()
print("delta test: C2[2,:4] =", C2[2, :4], " ref:", B2[:4, 5])
nz = C2.abs() > 1e-6
print("nonzero rows:", nz.any(1).nonzero().flatten().tolist())
