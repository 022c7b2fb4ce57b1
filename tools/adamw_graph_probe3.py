import sys, faulthandler, torch
faulthandler.enable()
sys.path.insert(0, "/root/repo")
import torch.nn as nn
from vantage6_amd.ops.gemm import Linear as MfmaLinear
from vantage6_amd.parallel.flat import FlatParamSpace
from vantage6_amd.ops import hip_ops
ops = hip_ops()
DEV = "cuda:0"
which = sys.argv[1]
torch.manual_seed(0)
net = nn.Sequential(MfmaLinear(64, 128), nn.ReLU(), MfmaLinear(128, 10)).to(DEV, torch.bfloat16)
flat = FlatParamSpace(net, scatter_grads=True)
x = torch.randn(16, 64, device=DEV, dtype=torch.bfloat16)
y = torch.randint(0, 10, (16,), device=DEV)

def fwdbwd():
    flat.zero_grad()
    loss = torch.nn.functional.cross_entropy(net(x).float(), y)
    loss.backward()
    return loss

def body():
    loss = fwdbwd()
    if which == "sgd":
        flat.sgd_step(lr=0.05)
    elif which == "adamw":
        flat.adamw_step(lr=1e-3)
    elif which == "adamw_norefresh":
        ops.fused_adamw_mt(flat._sgd_table, flat.param_flat, flat.master_flat,
                           flat.momentum_flat, flat.exp_avg_sq, 1e-3, 0.9,
                           0.999, 1e-8, 0.01, 1, flat._step_dev)
    elif which == "adamw_hostbc":
        flat.refresh_sgd_table()
        ops.fused_adamw_mt(flat._sgd_table, flat.param_flat, flat.master_flat,
                           flat.momentum_flat, flat.exp_avg_sq, 1e-3, 0.9,
                           0.999, 1e-8, 0.01, 1, None)
    return loss

for i in range(3):
    body()
torch.cuda.synchronize()
print("eager ok", flush=True)
side = torch.cuda.Stream(); side.wait_stream(torch.cuda.current_stream())
with torch.cuda.stream(side):
    for i in range(3):
        body()
torch.cuda.current_stream().wait_stream(side)
torch.cuda.synchronize(); print("warmup ok", flush=True)
g = torch.cuda.CUDAGraph()
with torch.cuda.graph(g):
    body()
print("captured", flush=True)
g.replay(); torch.cuda.synchronize()
print("PASS", which, flush=True)
