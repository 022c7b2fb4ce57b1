#!/usr/bin/env python3
"""FedAvg deep-learning round on the engine (single process).

CPU demo with logreg:   python examples/fedavg_training.py
On an MI355X node this same engine runs ResNet/BERT/Llama at one rank
per GPU over RCCL (see bench.py); enable DP-SGD or secure aggregation
with the flags below.
"""
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parents[1]))

import torch

from algorithms.fedavg import fit_rounds


def main() -> None:
    g = torch.Generator().manual_seed(0)
    x = torch.randn(512, 16, generator=g)
    w = torch.randn(16, 2, generator=g)
    dataset = {"x": x, "y": (x @ w).argmax(-1)}

    out = fit_rounds(
        dataset,
        model="logreg",
        model_kwargs={"in_features": 16, "num_classes": 2},
        rounds=5,
        local_steps=4,
        batch_size=64,
        lr=0.1,
        device=torch.device("cpu"),
        # fedprox_mu=0.01,          # FedProx proximal term
        # server_adam=True,         # FedAdam server optimizer
        # secure_agg=True,          # int64 pairwise-mask aggregation
        # dp={"max_grad_norm": 1.0, "noise_multiplier": 1.0},  # DP-SGD
    )
    for h in out["history"]:
        print(f"round {h['round']}: loss {h['loss']:.4f}")


if __name__ == "__main__":
    main()
