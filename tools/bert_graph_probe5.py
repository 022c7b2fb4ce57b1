import sys, torch
import torch.nn.functional as F
sys.path.insert(0, "/root/repo")
from vantage6_amd.models.factory import build_model, model_loss

mode = sys.argv[1]  # fwd | fwdbwd | fwdbwd_nosdpa | fwd_nosdpa
DEV = torch.device("cuda:0")
torch.manual_seed(0)

if "nosdpa" in mode:
    def manual_sdpa(q, k, v, is_causal=False, enable_gqa=False, **kw):
        s = (q.float() @ k.float().transpose(-1, -2)) / (q.shape[-1] ** 0.5)
        return (F.softmax(s, dim=-1) @ v.float()).to(q.dtype)
    F.scaled_dot_product_attention = manual_sdpa

net = build_model("bert_base").to(DEV, torch.bfloat16)
loss_fn = model_loss("bert_base")
g = torch.Generator().manual_seed(1)
S, B = 512, 8
def batch():
    ids = torch.randint(0, 30522, (B, S + 1), generator=g)
    return {"input_ids": ids[:, :-1].to(DEV), "labels": ids[:, 1:].to(DEV)}
static = {k: v.clone() for k, v in batch().items()}

def body():
    if "bwd" in mode:
        for p in net.parameters():
            p.grad = None
    loss = loss_fn(net, static)
    if "bwd" in mode:
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
