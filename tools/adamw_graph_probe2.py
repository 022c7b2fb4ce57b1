import sys, torch
sys.path.insert(0, "/root/repo")
from vantage6_amd.ops import hip_ops
ops = hip_ops()
DEV = "cuda:0"
step_dev = torch.zeros(1, dtype=torch.int32, device=DEV)
n = 1 << 16
param = torch.zeros(n, dtype=torch.bfloat16, device=DEV)
master = torch.zeros(n, dtype=torch.float32, device=DEV)
grad = torch.randn(n, dtype=torch.bfloat16, device=DEV)
ea = torch.zeros_like(master); es = torch.zeros_like(master)

print("eager fused_adamw with step_dev...", flush=True)
ops.fused_adamw(param, master, grad, ea, es, 1e-3, 0.9, 0.999, 1e-8, 0.01, 0, step_dev)
torch.cuda.synchronize(); print("ok, step_dev =", int(step_dev.item()), flush=True)

print("warmup capture stream...", flush=True)
side = torch.cuda.Stream(); side.wait_stream(torch.cuda.current_stream())
with torch.cuda.stream(side):
    ops.fused_adamw(param, master, grad, ea, es, 1e-3, 0.9, 0.999, 1e-8, 0.01, 0, step_dev)
torch.cuda.current_stream().wait_stream(side)
torch.cuda.synchronize(); print("ok", flush=True)

print("capture fused_adamw alone...", flush=True)
g = torch.cuda.CUDAGraph()
with torch.cuda.graph(g):
    ops.fused_adamw(param, master, grad, ea, es, 1e-3, 0.9, 0.999, 1e-8, 0.01, 0, step_dev)
print("captured", flush=True)
g.replay(); torch.cuda.synchronize()
print("replayed, step_dev =", int(step_dev.item()), flush=True)

print("capture WITHOUT step_dev (host bc)...", flush=True)
g2 = torch.cuda.CUDAGraph()
with torch.cuda.graph(g2):
    ops.fused_adamw(param, master, grad, ea, es, 1e-3, 0.9, 0.999, 1e-8, 0.01, 3, None)
print("captured2", flush=True)
g2.replay(); torch.cuda.synchronize(); print("done all", flush=True)
