"""DP-SGD ghost-clipping tests (CPU): ghost norms must equal brute-force
per-sample gradient norms, and the clipped gradient must equal Σ c_i g_i."""
import pytest
import torch
import torch.nn as nn

from vantage6_amd.ops.gemm import Linear as MfmaLinear
from vantage6_amd.ops.norms import LayerNorm
from vantage6_amd.parallel.dp import DPConfig, DPEngine


class TinyModel(nn.Module):
    """Embedding + LayerNorm + 2 Linears — every supported module kind."""

    def __init__(self, vocab=13, hidden=16, classes=3, seq=5):
        super().__init__()
        self.emb = nn.Embedding(vocab, hidden)
        self.ln = LayerNorm(hidden, dtype=torch.float32)
        self.fc1 = MfmaLinear(hidden, 24, dtype=torch.float32)
        self.fc2 = MfmaLinear(24, classes, dtype=torch.float32)

    def forward(self, ids):
        x = self.emb(ids)  # [B,S,H]
        x = self.ln(x)
        x = torch.relu(self.fc1(x))
        return self.fc2(x.mean(1))


def per_sample_losses(model, ids, y):
    logits = model(ids)
    return torch.nn.functional.cross_entropy(logits, y, reduction="none")


def brute_force_norms(model, ids, y):
    """One backward per sample — the ground truth the ghost trick must match."""
    B = ids.shape[0]
    norms = []
    grads = []
    for i in range(B):
        model.zero_grad()
        loss_i = per_sample_losses(model, ids[i : i + 1], y[i : i + 1])[0]
        loss_i.backward()
        g = torch.cat([p.grad.reshape(-1) for p in model.parameters()])
        norms.append(g.norm())
        grads.append(g.clone())
    model.zero_grad()
    return torch.stack(norms), torch.stack(grads)


@pytest.fixture
def setup():
    torch.manual_seed(0)
    model = TinyModel().float()
    B, S = 6, 5
    # include duplicate tokens within a sample (embedding scatter-norm case)
    ids = torch.tensor([[1, 2, 1, 3, 1], [4, 4, 4, 4, 4], [0, 1, 2, 3, 4],
                        [5, 6, 7, 8, 9], [2, 2, 3, 3, 4], [10, 11, 12, 0, 0]])
    y = torch.randint(0, 3, (B,))
    return model, ids, y


def test_ghost_norms_match_brute_force(setup):
    model, ids, y = setup
    ref_norms, _ = brute_force_norms(model, ids, y)

    engine = DPEngine(model, DPConfig(max_grad_norm=1.0))
    losses = per_sample_losses(model, ids, y)
    engine._norms = None
    engine._capturing = True
    model.zero_grad()
    losses.sum().backward()
    engine._capturing = False
    ghost = engine._norms.clamp_min(0).sqrt()
    assert torch.allclose(ghost, ref_norms, rtol=1e-4, atol=1e-5), (
        ghost, ref_norms)


def test_clipped_gradient_is_weighted_sum(setup):
    model, ids, y = setup
    C = 0.05  # force clipping of every sample
    ref_norms, ref_grads = brute_force_norms(model, ids, y)
    clip = (C / (ref_norms + 1e-6)).clamp(max=1.0)
    expect = (ref_grads * clip[:, None]).sum(0)

    engine = DPEngine(model, DPConfig(max_grad_norm=C))
    losses = per_sample_losses(model, ids, y)
    stats = engine.compute_clipped_grads(losses, model.zero_grad, ids.shape[0])
    got = torch.cat([p.grad.reshape(-1) for p in model.parameters()])
    assert stats["clipped_fraction"] == 1.0
    assert torch.allclose(got, expect, rtol=1e-4, atol=1e-5)


def test_unsupported_module_rejected():
    class Bad(nn.Module):
        def __init__(self):
            super().__init__()
            self.conv = nn.Conv2d(3, 4, 3)

    with pytest.raises(TypeError):
        DPEngine(Bad(), DPConfig())


def test_dp_fit_rounds_end_to_end():
    """fit_rounds with dp on a logreg: runs, loss finite, noise applied."""
    from algorithms.fedavg import fit_rounds
    from vantage6_amd.parallel.comm import CommContext

    g = torch.Generator().manual_seed(3)
    x = torch.randn(256, 16, generator=g)
    dataset = {"x": x, "y": (x.sum(-1) > 0).long()}
    out = fit_rounds(
        dataset, model="logreg", model_kwargs={"in_features": 16, "num_classes": 2},
        rounds=2, local_steps=2, batch_size=64, lr=0.05,
        dp={"max_grad_norm": 1.0, "noise_multiplier": 0.5},
        comm=CommContext(), device=torch.device("cpu"),
    )
    assert out["final_loss"] == out["final_loss"]  # not NaN


def test_epsilon_report():
    engine = DPEngine(nn.Sequential(nn.Linear(4, 2)), DPConfig(noise_multiplier=1.0))
    rep = engine.epsilon_report(steps=100, sample_rate=0.01)
    assert 0 < rep["epsilon"] < 100


def test_rdp_accountant_properties():
    """RDP epsilon: tighter than the loose bound at realistic settings,
    monotone in steps, decreasing in sigma, and for q=1/T=1 it matches the
    analytic Gaussian-mechanism minimum over alpha."""
    import math

    from vantage6_amd.parallel.dp import DPConfig, DPEngine
    import torch.nn as nn

    def eng(sigma):
        return DPEngine(nn.Linear(4, 2), DPConfig(noise_multiplier=sigma,
                                                  max_grad_norm=1.0,
                                                  delta=1e-5))
    e = eng(1.1)
    r = e.epsilon_report(steps=1000, sample_rate=0.01)
    assert r["epsilon"] < r["epsilon_loose"]
    assert 0.1 < r["epsilon"] < 5.0  # opacus reports ~0.9-1.1 here
    assert e.epsilon_report(2000, 0.01)["epsilon"] > r["epsilon"]
    assert eng(2.0).epsilon_report(1000, 0.01)["epsilon"] < r["epsilon"]
    # q=1, T=1: epsilon = min_alpha alpha/(2 sigma^2) + log(1/delta)/(alpha-1)
    sigma = 3.0
    got = eng(sigma).epsilon_report(1, 1.0)["epsilon"]
    want = min(a / (2 * sigma**2) + math.log(1e5) / (a - 1)
               for a in range(2, 1025))
    assert got == pytest.approx(want, rel=1e-6)
