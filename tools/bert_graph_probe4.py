import sys, torch
sys.path.insert(0, "/root/repo")
from vantage6_amd.models.factory import build_model, model_loss
from vantage6_amd.parallel.comm import CommContext
from vantage6_amd.parallel.fedavg import FedAvgConfig, FedAvgTrainer

mode = sys.argv[1]  # plain | refresh | noopt
DEV = torch.device("cuda:0")
torch.manual_seed(0)
net = build_model("bert_base").to(DEV, torch.bfloat16)
g = torch.Generator().manual_seed(1)
S, B = 512, 8
def batches():
    while True:
        ids = torch.randint(0, 30522, (B, S + 1), generator=g)
        yield {"input_ids": ids[:, :-1].to(DEV), "labels": ids[:, 1:].to(DEV)}
cfg = FedAvgConfig(local_steps=1, lr=(0.0 if mode == "noopt" else 1e-4),
                   optimizer="adamw")
tr = FedAvgTrainer(net, model_loss("bert_base"), batches(), CommContext(device=DEV),
                   cfg, n_samples=B)
for _ in range(3):
    tr.local_step()
assert tr.enable_graph()
names = [n for n, p in net.named_parameters() if p.requires_grad]
tbl0 = tr.flat._sgd_table.clone()
ptrs0 = [p.grad.data_ptr() for p, *_ in tr.flat._specs]
for i in range(6):
    tr.local_step()
    torch.cuda.synchronize()
    tbl_same = bool(torch.equal(tbl0, tr.flat._sgd_table))
    ptrs_same = ptrs0 == [p.grad.data_ptr() for p, *_ in tr.flat._specs]
    badg = [nm for nm, (p, *_ ) in zip(names, tr.flat._specs)
            if not torch.isfinite(p.grad.float()).all().item()][:3]
    mfin = bool(torch.isfinite(tr.flat.master_flat).all().item())
    print(f"replay {i}: loss {float(tr._static_loss.detach()):.4f} tbl_same {tbl_same} "
          f"ptrs_same {ptrs_same} master_fin {mfin} bad_grads {badg}", flush=True)
    if mode == "refresh":
        tr.flat.refresh_sgd_table(strict=True)
print("PASS", mode, flush=True)
