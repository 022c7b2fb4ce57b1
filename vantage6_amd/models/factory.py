"""Model + loss factory used by the fedavg algorithm packages and bench."""
from __future__ import annotations

from typing import Callable

import torch
import torch.nn.functional as F


def build_model(name: str, device="cpu", dtype=torch.bfloat16, **kw) -> torch.nn.Module:
    name = name.lower()
    if name in ("logreg", "logistic_regression"):
        from .logreg import logreg

        return logreg(device=device, dtype=dtype, **kw)
    if name in ("resnet18", "resnet-18"):
        from .resnet import resnet18

        return resnet18(device=device, dtype=dtype, **kw)
    if name in ("resnet50", "resnet-50"):
        from .resnet import resnet50

        return resnet50(device=device, dtype=dtype, **kw)
    if name in ("bert", "bert_base", "bert-base"):
        from .transformer import bert_base

        return bert_base(device=device, dtype=dtype, **kw)
    if name in ("llama", "llama_1b", "llama-1b", "llama-3-1b"):
        from .transformer import llama_1b

        return llama_1b(device=device, dtype=dtype, **kw)
    if name in ("vit", "vit_b16", "vit-b16"):
        from .transformer import vit_b16

        return vit_b16(device=device, dtype=dtype, **kw)
    raise ValueError(f"unknown model {name!r}")


def model_loss(name: str) -> Callable:
    """(model, batch) -> scalar loss, fp32 loss computation."""
    name = name.lower()
    if name in ("bert", "bert_base", "bert-base", "llama", "llama_1b", "llama-1b",
                "llama-3-1b"):

        def lm_loss(model, batch):
            from ..ops.cross_entropy import cross_entropy

            ids = batch["input_ids"]
            logits = model(ids[:, :-1])
            vl = getattr(model, "vocab_logical", logits.shape[-1])
            if vl < logits.shape[-1]:
                # padded LM head (BertBase vocab_pad): pad classes get
                # probability exactly 0 and exactly-zero grads
                logits = logits.clone()
                logits[..., vl:] = float("-inf")
            return cross_entropy(logits, ids[:, 1:])

        return lm_loss

    def cls_loss(model, batch):
        logits = model(batch["x"])
        return F.cross_entropy(logits.float(), batch["y"])

    return cls_loss


def model_loss_per_sample(name: str) -> Callable:
    """(model, batch) -> per-sample loss vector [B] (DP-SGD needs this)."""
    name = name.lower()
    if name in ("bert", "bert_base", "bert-base", "llama", "llama_1b", "llama-1b",
                "llama-3-1b"):

        def lm_loss_ps(model, batch):
            from ..ops.cross_entropy import cross_entropy

            ids = batch["input_ids"]
            logits = model(ids[:, :-1])
            losses = cross_entropy(logits, ids[:, 1:], reduction="none")
            return losses.reshape(ids.shape[0], -1).mean(-1)

        return lm_loss_ps

    def cls_loss_ps(model, batch):
        logits = model(batch["x"])
        return F.cross_entropy(logits.float(), batch["y"], reduction="none")

    return cls_loss_ps
