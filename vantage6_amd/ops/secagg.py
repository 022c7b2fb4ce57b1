"""Secure-aggregation masking (pairwise Philox masks, fixed-point int64).

Design (SURVEY.md §6.8 / §8.3 hard part #3): fp32 masks cannot cancel
bit-exactly under RCCL's unordered reduction, so the masked aggregate runs
in modular int64 arithmetic — quantize ``w_i * master`` to fixed point,
add, for every pair (i, j) with i < j, a full-range 64-bit mask with sign
+1 on rank i and −1 on rank j. All-reduce(SUM) wraps mod 2^64 ⇒ masks
cancel exactly; the server and every other station only ever see masked
integers. GPU path: one fused Philox kernel (csrc/secagg.hip); CPU path
below is the reference. The CPU and GPU paths use different mask PRNGs,
so a secure-aggregation job must be device-homogeneous across ranks (it
always is: one torchrun launch) — masks only cancel against themselves.
"""
from __future__ import annotations

import torch

from . import hip_ops


def _pair_seed(seed: int, i: int, j: int) -> int:
    # deterministic per (round-seed, pair); same value on both ranks
    return (seed * 0x9E3779B97F4A7C15 + (i << 20) + j) % (2**63)


def quantize_mask(
    q_out: torch.Tensor,
    master: torch.Tensor,
    weight: float,
    scale: float,
    *,
    seed: int,
    rank: int,
    world_size: int,
) -> None:
    """q = round(weight*master*scale) + Σ_{j>rank} m(rank,j) − Σ_{j<rank} m(j,rank)."""
    if master.is_cuda:
        hip_ops().secagg_quantize_mask(
            q_out, master, weight, scale, seed, rank, world_size
        )
        return
    q_out.copy_((master.double() * weight * scale).round().to(torch.int64))
    n = q_out.numel()
    for j in range(world_size):
        if j == rank:
            continue
        lo_rank, hi_rank = min(rank, j), max(rank, j)
        g = torch.Generator().manual_seed(_pair_seed(seed, lo_rank, hi_rank))
        hi = torch.randint(0, 2**32, (n,), generator=g, dtype=torch.int64)
        lo = torch.randint(0, 2**32, (n,), generator=g, dtype=torch.int64)
        mask = (hi << 32) | lo
        if rank == lo_rank:
            q_out.add_(mask)
        else:
            q_out.sub_(mask)
