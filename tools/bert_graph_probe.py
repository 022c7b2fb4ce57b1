import sys, torch
sys.path.insert(0, "/root/repo")
from vantage6_amd.models.factory import build_model, model_loss
from vantage6_amd.parallel.comm import CommContext
from vantage6_amd.parallel.fedavg import FedAvgConfig, FedAvgTrainer

mode = sys.argv[1] if len(sys.argv) > 1 else "full"
DEV = torch.device("cuda:0")
torch.manual_seed(0)
net = build_model("bert_base").to(DEV, torch.bfloat16)
if mode == "freeze_emb":
    for m in net.modules():
        if isinstance(m, torch.nn.Embedding):
            for p in m.parameters():
                p.requires_grad_(False)
if mode == "freeze_bias":
    for n, p in net.named_parameters():
        if n.endswith("bias"):
            p.requires_grad_(False)
g = torch.Generator().manual_seed(1)
S, B = 512, 8
def batches():
    while True:
        ids = torch.randint(0, 30522, (B, S + 1), generator=g)
        yield {"input_ids": ids[:, :-1].to(DEV), "labels": ids[:, 1:].to(DEV)}
tr = FedAvgTrainer(net, model_loss("bert_base"), batches(), CommContext(device=DEV),
                   FedAvgConfig(local_steps=1, lr=1e-4, optimizer="adamw"),
                   n_samples=B)
for _ in range(3):
    tr.local_step()
ok = tr.enable_graph()
print("graph:", ok, flush=True)
for i in range(8):
    tr.local_step()
    torch.cuda.synchronize()
    loss = float(tr._static_loss.detach())
    mfin = bool(torch.isfinite(tr.flat.master_flat).all().item())
    print(f"replay {i}: loss {loss:.4f} master_finite {mfin}", flush=True)
print("PASS", mode, flush=True)
