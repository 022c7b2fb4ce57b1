import sys, time, torch
sys.path.insert(0, "/root/repo")
from vantage6_amd.ops import hip_ops
ops = hip_ops()
def t(fn, n=50):
    for _ in range(10): fn()
    torch.cuda.synchronize(); t0=time.perf_counter()
    for _ in range(n): fn()
    torch.cuda.synchronize(); return (time.perf_counter()-t0)/n*1e6
for m,nn in [(4096,768),(4096,3072),(4096,2304),(32,1000)]:
    dy = torch.randn(m,nn,dtype=torch.bfloat16,device="cuda")
    us_ours = t(lambda: ops.colsum_bf16(dy))
    us_torch = t(lambda: dy.sum(0))
    rel = (ops.colsum_bf16(dy).float()-dy.float().sum(0)).abs().max().item()
    print(f"[{m}x{nn}] ours {us_ours:.1f}us torch {us_torch:.1f}us maxerr {rel:.4f}")
