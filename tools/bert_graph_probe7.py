import sys, functools, torch
import torch.nn as nn
import torch.nn.functional as F
sys.path.insert(0, "/root/repo")
import vantage6_amd.models.transformer as T

mode = sys.argv[1]  # torchln | nolinbias | relu
if mode == "torchln":
    T.LayerNorm = lambda h: nn.LayerNorm(h)
if mode == "nolinbias":
    _ML = T.MfmaLinear
    T.MfmaLinear = functools.partial(_ML, bias=False)
if mode == "relu":
    F.gelu = torch.relu
DEV = torch.device("cuda:0")
torch.manual_seed(0)
net = T.BertBase(layers=2).to(DEV, torch.bfloat16)
g = torch.Generator().manual_seed(1)
S, B = 512, 8
def batch():
    ids = torch.randint(0, 30522, (B, S + 1), generator=g)
    return {"input_ids": ids[:, :-1].to(DEV), "labels": ids[:, 1:].to(DEV)}
static = {k: v.clone() for k, v in batch().items()}
def body():
    for p in net.parameters():
        p.grad = None
    out = net(static["input_ids"])
    loss = torch.nn.functional.cross_entropy(
        out.reshape(-1, 30522).float(), static["labels"].reshape(-1))
    loss.backward()
    return loss
for _ in range(3):
    body()
torch.cuda.synchronize()
side = torch.cuda.Stream(); side.wait_stream(torch.cuda.current_stream())
with torch.cuda.stream(side):
    for _ in range(3):
        body()
torch.cuda.current_stream().wait_stream(side)
gph = torch.cuda.CUDAGraph()
with torch.cuda.graph(gph):
    sl = body()
torch.cuda.synchronize()
for i in range(6):
    for k, v in batch().items():
        static[k].copy_(v)
    gph.replay()
    torch.cuda.synchronize()
    print(f"replay {i}: loss {float(sl.detach()):.4f}", flush=True)
print("PASS", mode, flush=True)
