"""DP-SGD: per-sample gradient clipping (ghost norms) + gaussian noise.

SURVEY.md §8.3 hard part #2 — per-sample clipping at BERT scale without
materializing per-sample gradients:

1. forward once → per-sample losses L[B],
2. backward #1 (sum L, retain graph): module hooks compute each sample's
   TOTAL gradient norm² from (saved input, grad_output) pairs — the ghost
   norm trick: for a Linear, ||dW_i||² = ⟨x_i x_iᵀ, dy_i dy_iᵀ⟩, O(T²(K+N))
   instead of O(N·K) memory,
3. clip factors c_i = min(1, C/||g_i||),
4. backward #2 of Σ c_i L_i → the clipped gradient SUM lands in the flat
   grad buffer,
5. one fused kernel pass (ops/csrc/elementwise.hip::dp_scale_noise_kernel)
   rescales by 1/B and adds N(0, σC/B) Philox noise.

Supported modules: ops.gemm.Linear, nn.Linear, ops.norms.LayerNorm,
nn.LayerNorm, nn.Embedding. A model containing any *other* parameterized
module raises (silent non-private training is worse than an error).
"""
from __future__ import annotations

import math
from dataclasses import dataclass
from typing import Optional

import torch
import torch.nn as nn


@dataclass
class DPConfig:
    max_grad_norm: float = 1.0  # C
    noise_multiplier: float = 1.0  # sigma
    # None (the default, and the only safe production value) → each DPEngine
    # draws its noise seed from the OS CSPRNG, so no other party — including
    # sibling stations running the same config dict — can predict or cancel
    # the noise. An explicit int is for reproducible unit tests ONLY, and even
    # then the engine decorrelates ranks by mixing the distributed rank in.
    seed: Optional[int] = None
    delta: float = 1e-5


_SUPPORTED = "Linear/LayerNorm/Embedding"


class DPEngine:
    def __init__(self, model: nn.Module, cfg: DPConfig):
        self.model = model
        self.cfg = cfg
        if cfg.seed is None:
            import secrets

            self._noise_seed = secrets.randbits(62)
        else:
            # test-only deterministic path; still never bit-identical across
            # ranks (identical noise would cancel in the FedAvg difference)
            rank = 0
            if torch.distributed.is_available() and torch.distributed.is_initialized():
                rank = torch.distributed.get_rank()
            self._noise_seed = (cfg.seed + rank * 0x9E3779B97F4A7C15) & ((1 << 62) - 1)
        self._norms: Optional[torch.Tensor] = None  # [B] accumulating norm^2
        self._capturing = False
        self._step = 0
        self._step_dev: Optional[torch.Tensor] = None  # int32 device counter
        self._stats_dev: Optional[torch.Tensor] = None  # [median, max, frac]
        self._hooked = []
        for mod in model.modules():
            has_params = any(True for _ in mod.parameters(recurse=False))
            if not has_params:
                continue
            from ..ops.gemm import Linear as MfmaLinear
            from ..ops.norms import LayerNorm as HipLayerNorm

            if isinstance(mod, (MfmaLinear, nn.Linear)):
                kind = "linear"
            elif isinstance(mod, (HipLayerNorm, nn.LayerNorm)):
                kind = "layernorm"
            elif isinstance(mod, nn.Embedding):
                kind = "embedding"
            else:
                raise TypeError(
                    f"DP-SGD: unsupported parameterized module {type(mod).__name__} "
                    f"(supported: {_SUPPORTED})"
                )
            mod.register_forward_pre_hook(self._save_input)
            mod.register_full_backward_hook(self._make_bwd_hook(kind))
            self._hooked.append(mod)

    # ------------------------------------------------------------- hooks
    def _save_input(self, mod, inputs):
        # always save (forward runs before the capture flag is set); the
        # tensor is already retained by the autograd graph, this is a ref
        mod._dp_input = inputs[0].detach()

    def _make_bwd_hook(self, kind):
        def hook(mod, grad_input, grad_output):
            if not self._capturing:
                return
            dy = grad_output[0].detach()
            x = mod._dp_input
            self._accumulate(kind, mod, x, dy)

        return hook

    def _acc(self, n2: torch.Tensor):
        if self._norms is None:
            self._norms = n2.float()
        else:
            self._norms = self._norms + n2.float()

    def _accumulate(self, kind, mod, x, dy):
        B = dy.shape[0]
        if kind == "linear":
            xf = x.reshape(B, -1, x.shape[-1]).float()  # [B,T,K]
            dyf = dy.reshape(B, -1, dy.shape[-1]).float()  # [B,T,N]
            T = xf.shape[1]
            if T == 1:
                n2 = xf.pow(2).sum((1, 2)) * dyf.pow(2).sum((1, 2))
            else:
                # ghost norm: <x x^T, dy dy^T> per sample
                xxt = torch.bmm(xf, xf.transpose(1, 2))  # [B,T,T]
                yyt = torch.bmm(dyf, dyf.transpose(1, 2))
                n2 = (xxt * yyt).sum((1, 2))
            if getattr(mod, "bias", None) is not None:
                n2 = n2 + dyf.sum(1).pow(2).sum(-1)
            self._acc(n2)
        elif kind == "layernorm":
            xf = x.reshape(B, -1, x.shape[-1]).float()
            dyf = dy.reshape(B, -1, dy.shape[-1]).float()
            mean = xf.mean(-1, keepdim=True)
            var = xf.var(-1, unbiased=False, keepdim=True)
            eps = getattr(mod, "eps", 1e-5)
            xhat = (xf - mean) * (var + eps).rsqrt()
            dw_i = (dyf * xhat).sum(1)  # [B,H]
            n2 = dw_i.pow(2).sum(-1)
            if getattr(mod, "bias", None) is not None:
                n2 = n2 + dyf.sum(1).pow(2).sum(-1)
            self._acc(n2)
        elif kind == "embedding":
            ids = x.reshape(B, -1)
            dyf = dy.reshape(B, ids.shape[1], -1).float()
            V, H = mod.weight.shape
            buf = torch.zeros(V, H, device=dy.device, dtype=torch.float32)
            n2 = torch.empty(B, device=dy.device)
            for i in range(B):  # exact per-sample scatter norm (dup tokens sum)
                buf.zero_()
                buf.index_add_(0, ids[i], dyf[i])
                n2[i] = buf.pow(2).sum()
            self._acc(n2)

    # -------------------------------------------------------------- step
    def compute_clipped_grads(self, per_sample_losses: torch.Tensor,
                              zero_grad_fn, batch_size: int,
                              sync_stats: bool = True) -> dict:
        """Run the two-pass ghost-clipping protocol; leaves the clipped,
        noised MEAN gradient in the parameters' .grad buffers.

        ``sync_stats=False`` is the hipGraph-capturable form: every op is
        device-side; per-step stats land in a persistent device buffer
        (read them with :meth:`stats`) instead of forcing a host sync."""
        cfg = self.cfg
        B = per_sample_losses.shape[0]
        # pass 1: norms
        self._norms = None
        self._capturing = True
        zero_grad_fn()
        per_sample_losses.sum().backward(retain_graph=True)
        self._capturing = False
        norms = self._norms.clamp_min(0).sqrt()  # [B]
        clip = (cfg.max_grad_norm / (norms + 1e-6)).clamp(max=1.0)
        # pass 2: clipped sum
        zero_grad_fn()
        (per_sample_losses * clip.detach()).sum().backward()
        self._step += 1
        if self._stats_dev is None or self._stats_dev.device != norms.device:
            # allocated during eager warmup (i.e. BEFORE any graph capture)
            # so it is a persistent buffer, not a capture-pool tensor
            self._stats_dev = torch.zeros(3, device=norms.device)
        self._stats_dev[0].copy_(norms.median())
        self._stats_dev[1].copy_(norms.max())
        self._stats_dev[2].copy_((clip < 1.0).float().mean())
        if not sync_stats:
            return {}
        return self.stats()

    def stats(self) -> dict:
        """Read the last step's clip stats (host sync; call off the hot
        path — under graph replay the device buffer is updated per replay)."""
        if self._stats_dev is None:
            return {}
        med, mx, frac = self._stats_dev.tolist()
        return {"median_grad_norm": med, "max_grad_norm": mx,
                "clipped_fraction": frac}

    def add_noise_and_scale(self, grad_flat: torch.Tensor, batch_size: int) -> None:
        """grad = grad/B + N(0, sigma*C/B). One fused pass on GPU; the
        per-step seed component comes from an int32 DEVICE counter so the
        pass is hipGraph-capturable (each replay draws fresh noise)."""
        cfg = self.cfg
        std = cfg.noise_multiplier * cfg.max_grad_norm / batch_size
        if grad_flat.is_cuda:
            from ..ops import hip_ops

            if self._step_dev is None:
                self._step_dev = torch.zeros(1, dtype=torch.int32,
                                             device=grad_flat.device)
            hip_ops().dp_scale_noise(grad_flat, 1.0 / batch_size, std,
                                     self._noise_seed, self._step_dev)
        else:
            g = torch.Generator().manual_seed(self._noise_seed + self._step)
            noise = torch.randn(grad_flat.shape, generator=g) * std
            grad_flat.mul_(1.0 / batch_size).add_(noise.to(grad_flat.dtype))

    def epsilon_report(self, steps: int, sample_rate: float) -> dict:
        """(eps, delta) via Renyi-DP accounting of the subsampled Gaussian
        mechanism (Mironov 2017; the accounting opacus uses): integer-alpha
        RDP of one step is

            RDP(a) = log( sum_{k=0..a} C(a,k) (1-q)^(a-k) q^k
                          * exp((k^2-k)/(2 sigma^2)) ) / (a-1),

        composed linearly over ``steps`` and converted with
        eps = min_a [ T*RDP(a) + log(1/delta)/(a-1) ]. ``epsilon_loose``
        keeps the previous strong-composition bound for comparison."""
        q, sigma, delta = sample_rate, self.cfg.noise_multiplier, self.cfg.delta
        if sigma <= 0:
            return {"epsilon": float("inf"), "delta": delta}
        loose = math.sqrt(2 * steps * math.log(1 / delta)) * q / sigma + \
            steps * q * (math.exp(1.0 / sigma**2) - 1)
        eps = min(steps * self._rdp_step(q, sigma, a)
                  + math.log(1.0 / delta) / (a - 1)
                  for a in self._ALPHAS)
        return {"epsilon": eps, "epsilon_loose": loose, "delta": delta,
                "sigma": sigma, "steps": steps, "sample_rate": q}

    _ALPHAS = list(range(2, 65)) + [96, 128, 256, 512, 1024]

    @staticmethod
    def _rdp_step(q: float, sigma: float, alpha: int) -> float:
        """Integer-alpha RDP of one subsampled-Gaussian step (log-sum-exp)."""
        if q >= 1.0:
            return alpha / (2.0 * sigma * sigma)
        terms = []
        for k in range(alpha + 1):
            log_binom = (math.lgamma(alpha + 1) - math.lgamma(k + 1)
                         - math.lgamma(alpha - k + 1))
            terms.append(log_binom + (alpha - k) * math.log1p(-q)
                         + (k * math.log(q) if k else 0.0)
                         + (k * k - k) / (2.0 * sigma * sigma))
        m = max(terms)
        return (m + math.log(sum(math.exp(t - m) for t in terms))) / (alpha - 1)
