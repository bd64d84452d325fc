import os, sys, torch
sys.path.insert(0, ".")
os.environ["MI355_GEMM_TILE"] = "256"
from mi355 import ops
a = torch.randn(8192, 768, device="cuda", dtype=torch.bfloat16)
b = torch.randn(4096, 768, device="cuda", dtype=torch.bfloat16)
for _ in range(3):
    ops.gemm_nt(a, b)
torch.cuda.synchronize()
for _ in range(20):
    ops.gemm_nt(a, b)
torch.cuda.synchronize()
