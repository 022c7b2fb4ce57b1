"""Logistic regression (config 1) — a single MFMA Linear + CE loss."""
from __future__ import annotations

import torch

from ..ops.gemm import Linear as MfmaLinear


class LogisticRegression(torch.nn.Module):
    def __init__(self, in_features: int, num_classes: int):
        super().__init__()
        self.fc = MfmaLinear(in_features, num_classes)

    def forward(self, x):
        return self.fc(x)


def logreg(in_features: int = 128, num_classes: int = 2,
           dtype=torch.bfloat16, device="cpu") -> LogisticRegression:
    return LogisticRegression(in_features, num_classes).to(device=device, dtype=dtype)
