"""Distributed/data plane — RCCL over xGMI.

Reference parity note: the reference has NO
collective-communication code — its "aggregation" is a REST fan-out/poll
loop run by a user algorithm. Here aggregation is first-class: one process
per GPU (``torch.distributed`` backend "nccl" = RCCL on ROCm), flat fp32
parameter buffers, weighted FedAvg as bucketed all-reduce (or reduce +
broadcast) over the fully-connected 7-link xGMI topology, overlapped with
compute on dedicated HIP streams.
"""
from .comm import CommContext, init_from_env  # noqa: F401
from .fedavg import FedAvgConfig, FedAvgTrainer  # noqa: F401
from .flat import FlatParamSpace  # noqa: F401
