"""vantage6_amd — MI355X-native federated-learning infrastructure.

A ground-up re-design of the capabilities of the vantage6 federated-learning
framework (reference: IKNL/vantage6 monorepo; see SURVEY.md for the capability
inventory) for a single 8×MI355X node:

- the vantage6 *server* (Flask REST + Socket.IO + SQL task broker) becomes an
  in-process orchestrator with a FastAPI façade speaking the same JSON schema
  (``vantage6_amd.server``),
- each vantage6 *node* ("data station" daemon running algorithm Docker images)
  becomes a station worker pinned to one MI355X GPU executing algorithm
  packages in-process (``vantage6_amd.node``),
- the per-station compute (delegated to user Docker images in the reference)
  is first-class: hand-written CDNA4 HIP kernels — MFMA/LDS-tiled GEMM,
  LayerNorm/RMSNorm, fused SGD/AdamW, DP-SGD clip+noise, secure-aggregation
  masks (``vantage6_amd.ops``),
- FedAvg aggregation (a REST fan-out/poll loop in the reference) becomes RCCL
  reduce + broadcast over xGMI, overlapped with the next round's forward on
  HIP streams (``vantage6_amd.parallel``).

Reference parity citations in docstrings use the upstream monorepo layout
(``vantage6-<pkg>/vantage6/...``) as catalogued in SURVEY.md §3.
"""

__version__ = "0.1.0"
