#!/bin/bash
# Round-3 opening GPU call: measure everything round 2 built but could
# not measure (GPU budget exhausted after the close-out sweep).
# Usage: /usr/local/graft/bin/gpurun --timeout 900 -- 'bash tools/r3_first_measurements.sh'
set -x
cd "$GRAFT_REPO_ROOT" || exit 1
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out

# 1. full GPU suite (incl. the 5 fused-CE tests added late in r2)
timeout 400 python -m pytest tests -m gpu -x -q 2>&1 | tail -3

# 2. new model families on the HIP path (never GPU-measured in r2)
timeout 200 python bench.py --model resnet50 --steps 15 --warmup 5 2>/dev/null | tail -1 \
  | tee gpurun_out/bench_resnet50_r3.json
timeout 200 python bench.py --model vit_b16 --steps 10 --warmup 3 2>/dev/null | tail -1 \
  | tee gpurun_out/bench_vit_r3.json
# 3. BERT with the padded-vocab fused CE (r2 lead: NOTES.md item 5) —
#    edit models/transformer.py bert_base default or pass kwargs via a
#    micro-script, then compare against the 17.0 r/s baseline:
timeout 300 python - << 'EOF' 2>/dev/null | tail -1 | tee gpurun_out/bench_bert_vocabpad_r3.txt
# quick A/B: bert_base(vocab_pad=30720) engages ce_fwd/ce_bwd
import time

import torch

from vantage6_amd.models.factory import model_loss
from vantage6_amd.models.transformer import bert_base

torch.manual_seed(0)
m = bert_base(device="cuda", vocab_pad=30720)
loss_fn = model_loss("bert_base")
ids = torch.randint(0, 30522, (8, 513), device="cuda")
opt = torch.optim.SGD(m.parameters(), lr=0.0)
for _ in range(5):
    loss = loss_fn(m, {"input_ids": ids})
    loss.backward()
    opt.zero_grad(set_to_none=True)
torch.cuda.synchronize()
t0 = time.perf_counter()
for _ in range(20):
    loss = loss_fn(m, {"input_ids": ids})
    loss.backward()
    opt.zero_grad(set_to_none=True)
torch.cuda.synchronize()
print(f"bert vocab_pad fwd+bwd: {(time.perf_counter()-t0)/20*1e3:.2f} ms/step")
EOF

# 4. fresh rocprof for the resnet headline (conv wgrad is the top r3 lead)
cd /tmp && export TMPDIR=/tmp
timeout 300 bash -c "rocprofv3 --kernel-trace --stats -d /tmp/r3 -o rn -- \
  python $GRAFT_REPO_ROOT/bench.py --steps 20 --warmup 5 > /tmp/rn.json 2>/dev/null"
cd "$GRAFT_REPO_ROOT"
python tools/summarize_profile.py "/tmp/r3/*rn*.db" 600 \
  gpurun_out/resnet18_steady_r3_open.txt "# ResNet-18 r3 opening profile" | head -12
