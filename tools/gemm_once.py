import sys, torch
sys.path.insert(0, "/root/repo")
from vantage6_amd.ops import hip_ops
ops = hip_ops()
m = n = k = 4096
a = torch.randn(m, k, dtype=torch.bfloat16, device="cuda")
b = torch.randn(n, k, dtype=torch.bfloat16, device="cuda")
for _ in range(10):
    c = ops.gemm_bf16(a, b, False, True)
torch.cuda.synchronize()
