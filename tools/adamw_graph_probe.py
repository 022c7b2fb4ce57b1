import sys, torch
sys.path.insert(0, "/root/repo")
import torch.nn as nn
from vantage6_amd.ops.gemm import Linear as MfmaLinear
from vantage6_amd.ops.norms import LayerNorm
from vantage6_amd.parallel.flat import FlatParamSpace

DEV = "cuda:0"
torch.manual_seed(0)
net = nn.Sequential(MfmaLinear(64, 128), nn.ReLU(), MfmaLinear(128, 10)).to(DEV, torch.bfloat16)
flat = FlatParamSpace(net, scatter_grads=True)
x = torch.randn(16, 64, device=DEV, dtype=torch.bfloat16)
y = torch.randint(0, 10, (16,), device=DEV)

def step():
    flat.zero_grad()
    loss = torch.nn.functional.cross_entropy(net(x).float(), y)
    loss.backward()
    flat.adamw_step(lr=1e-3)
    return loss

print("eager step 1...", flush=True)
l = step(); torch.cuda.synchronize(); print("ok loss", float(l), "step_dev", int(flat._step_dev.item()), flush=True)
print("eager step 2...", flush=True)
l = step(); torch.cuda.synchronize(); print("ok", float(l), int(flat._step_dev.item()), flush=True)
print("warmup on side stream...", flush=True)
side = torch.cuda.Stream(); side.wait_stream(torch.cuda.current_stream())
with torch.cuda.stream(side):
    for i in range(3):
        step()
torch.cuda.current_stream().wait_stream(side)
torch.cuda.synchronize(); print("warmup done", int(flat._step_dev.item()), flush=True)
print("capture...", flush=True)
g = torch.cuda.CUDAGraph()
with torch.cuda.graph(g):
    sl = step()
print("captured", flush=True)
flat.refresh_sgd_table(strict=True)
torch.cuda.synchronize()
print("replay...", flush=True)
for i in range(3):
    g.replay()
torch.cuda.synchronize()
print("replayed, step_dev =", int(flat._step_dev.item()), "loss", float(sl), flush=True)
