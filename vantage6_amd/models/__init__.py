"""Model zoo for the federated benchmark configs (BASELINE.json):

- ``logreg``    — federated logistic regression (config 1)
- ``resnet18``  — the headline FedAvg model (config 2)
- ``bert_base`` — DP-SGD config (config 3)
- ``llama_1b``  — secure-aggregation config (config 4)

All models run in bf16 with our hand-written ops on the hot path
(ops.gemm.Linear, ops.norms.{LayerNorm,RMSNorm}); the reference framework
ships no models at all (compute lives in user Docker images),
so this zoo is part of the net-new MI355X compute plane.
"""
from .factory import build_model, model_loss  # noqa: F401
