import sys, json
from pathlib import Path
sys.path.insert(0, str(Path(__file__).resolve().parents[1]))
import torch
import torch.nn.functional as F
from vantage6_amd.ops import hip_ops
ops = hip_ops()
dev = "cuda:0"
torch.manual_seed(0)
for S in (64, 128, 192, 256, 512):
    for causal in (False, True):
        q = torch.randn(1, 1, S, 64, device=dev, dtype=torch.bfloat16)
        k = torch.randn(1, 1, S, 64, device=dev, dtype=torch.bfloat16)
        v = torch.randn(1, 1, S, 64, device=dev, dtype=torch.bfloat16)
        ref = F.scaled_dot_product_attention(q.float(), k.float(), v.float(),
                                             is_causal=causal)
        o, lse = ops.attn_fwd(q, k, v, causal, 64 ** -0.5, True)
        d = (o.float() - ref).abs()
        # where is it wrong? report per 16-row block
        rows_bad = d.amax(dim=-1)[0, 0]
        bad_blocks = [i for i in range(S // 16)
                      if rows_bad[i * 16:(i + 1) * 16].max().item() > 0.05
                      or rows_bad[i * 16:(i + 1) * 16].isnan().any()]
        print(json.dumps({"S": S, "causal": causal,
                          "max_err": round(d.max().item(), 4)
                          if not d.isnan().any() else "nan",
                          "bad_16row_blocks": bad_blocks[:12]}))
