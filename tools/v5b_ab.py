import json, sys, time
sys.path.insert(0, ".")
import torch
from vantage6_amd.ops import hip_ops
ops = hip_ops()
dev = "cuda:0"
torch.manual_seed(0)
# refcheck
a = torch.rand(512, 512, device=dev, dtype=torch.bfloat16) - 0.5
b = torch.rand(512, 512, device=dev, dtype=torch.bfloat16) - 0.5
ref = (a.float() @ b.float().t()).to(torch.bfloat16).float()
got = ops.gemm_nt_probe(a, b, 55, 1).float()
assert (got - ref).abs().max().item() < 0.05, "v5b WRONG"
print("refcheck ok")
# race screen
aa = torch.randn(2048, 768, device=dev, dtype=torch.bfloat16)
bb = torch.randn(2048, 768, device=dev, dtype=torch.bfloat16)
first = ops.gemm_nt_probe(aa, bb, 55, 1)
for _ in range(9):
    assert torch.equal(first, ops.gemm_nt_probe(aa, bb, 55, 1)), "RACE"
print("race ok")
for name, m, n, k in [("bert_qkv",4096,2304,768),("bert_fc1",4096,3072,768),
                      ("bert_fc2",4096,768,3072),("bert_proj",4096,768,768),
                      ("llama_qo",1024,2048,2048),("llama_gate",1024,8192,2048),
                      ("4k3",4096,4096,4096)]:
    a = torch.randn(m, k, device=dev, dtype=torch.bfloat16)
    b = torch.randn(n, k, device=dev, dtype=torch.bfloat16)
    fl = 2.0*m*n*k
    res = {}
    for v in (5, 55):
        for _ in range(3): ops.gemm_nt_probe(a, b, v, 1)
        torch.cuda.synchronize(); t0 = time.perf_counter()
        for _ in range(30): ops.gemm_nt_probe(a, b, v, 1)
        torch.cuda.synchronize()
        res[v] = round(fl/((time.perf_counter()-t0)/30)/1e12, 1)
    print(json.dumps({"shape": name, "v5": res[5], "v5b": res[55]}))
