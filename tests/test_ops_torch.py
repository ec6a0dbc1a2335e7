"""torch_ref op tests: the reference implementations verified against the
reference repo's own torch formulations (the equations of SURVEY §2.3)."""

import math

import pytest
import torch
from torch.distributions import Normal

from distributed_sac_amd.ops import torch_ref as R


def test_squashed_gaussian_matches_normal_formulation():
    torch.manual_seed(0)
    B, A, k = 64, 4, 1.0
    mu = torch.randn(B, A)
    log_std_raw = torch.randn(B, A) * 3  # exercise the clamp
    eps = torch.randn(B, A)

    action, logp, log_std = R.squashed_gaussian(mu, log_std_raw, eps, k)

    # reference formulation (LunarLander…/src/model.py:51-59)
    std = torch.exp(torch.clamp(log_std_raw, -20, 2))
    u = mu + std * eps
    ref_action = k * torch.tanh(u)
    normal = Normal(mu, std)
    glp = normal.log_prob(u)
    ref_logp = (glp - torch.log(k * (1 - (ref_action / k) ** 2 + 1e-6))
                ).sum(dim=-1, keepdim=True)

    assert torch.allclose(action, ref_action, atol=1e-6)
    assert torch.allclose(logp, ref_logp, atol=1e-5)
    assert torch.allclose(log_std, torch.clamp(log_std_raw, -20, 2))


def test_squashed_gaussian_k_scaling():
    torch.manual_seed(1)
    mu = torch.randn(8, 2)
    ls = torch.zeros(8, 2)
    eps = torch.randn(8, 2)
    k = 2.5
    a, logp, _ = R.squashed_gaussian(mu, ls, eps, k)
    assert a.abs().max() <= k
    std = torch.ones_like(mu)
    u = mu + std * eps
    ref = (Normal(mu, std).log_prob(u)
           - torch.log(k * (1 - torch.tanh(u) ** 2 + 1e-6))).sum(-1, keepdim=True)
    assert torch.allclose(logp, ref, atol=1e-5)


def test_td_target():
    torch.manual_seed(2)
    B = 32
    r = torch.randn(B, 1)
    d = (torch.rand(B, 1) < 0.3).float()
    q1, q2 = torch.randn(B, 1), torch.randn(B, 1)
    lp = torch.randn(B, 1)
    alpha = torch.tensor(0.2)
    y = R.td_target(r, d, q1, q2, lp, alpha, 0.99, 1.5)
    ref = 1.5 * r + 0.99 * (1 - d) * (torch.min(q1, q2) - 0.2 * lp)
    assert torch.allclose(y, ref, atol=1e-6)


def test_task_weights_and_alpha_gather():
    torch.manual_seed(3)
    B, T = 40, 10
    idx = torch.randint(0, T, (B,))
    one_hots = torch.nn.functional.one_hot(idx, T).float()
    alphas = torch.rand(T)
    w = R.task_weights(one_hots, alphas)
    assert torch.allclose(w.sum(), torch.tensor(1.0), atol=1e-6)
    soft = torch.softmax(-alphas, 0)
    raw = soft[idx]
    assert torch.allclose(w, raw / raw.sum(), atol=1e-6)

    log_alpha = torch.randn(T)
    g = R.gather_log_alpha(one_hots, log_alpha)
    assert g.shape == (B, 1)
    assert torch.allclose(g.squeeze(1), log_alpha[idx], atol=1e-6)


def test_entropy_from_log_std():
    torch.manual_seed(4)
    ls = torch.randn(16, 4)
    ent = R.entropy_from_log_std(ls)
    ref = (0.5 * 4 * (1 + math.log(2 * math.pi)) + ls.sum(-1)).mean()
    assert torch.allclose(ent, ref, atol=1e-6)


def test_polyak():
    torch.manual_seed(5)
    t = [torch.randn(3, 3), torch.randn(5)]
    s = [torch.randn(3, 3), torch.randn(5)]
    t0 = [x.clone() for x in t]
    R.polyak_(t, s, 0.1)
    for a, b, c in zip(t, s, t0):
        assert torch.allclose(a, 0.1 * b + 0.9 * c, atol=1e-6)
    R.polyak_(t, s, 1.0)
    for a, b in zip(t, s):
        assert torch.allclose(a, b)


def test_mlp_forward_matches_sequential():
    torch.manual_seed(6)
    import torch.nn as nn
    seq = nn.Sequential(nn.Linear(8, 32), nn.ReLU(), nn.Linear(32, 32),
                        nn.ReLU(), nn.Linear(32, 4))
    x = torch.randn(16, 8)
    ws = [m.weight for m in seq if isinstance(m, nn.Linear)]
    bs = [m.bias for m in seq if isinstance(m, nn.Linear)]
    assert torch.allclose(R.mlp_forward(x, ws, bs), seq(x), atol=1e-6)


def test_batched_linear_and_attention_pool():
    torch.manual_seed(7)
    k, B, i, o = 6, 12, 10, 5
    w = torch.randn(k, i, o)
    b = torch.randn(k, 1, o)
    x = torch.randn(B, i)
    y = R.batched_linear(x, w, b)
    assert y.shape == (k, B, o)
    ref = torch.einsum("kio,bi->kbo", w, x) + b
    assert torch.allclose(y, ref, atol=1e-6)
    y2 = R.batched_linear(y, torch.randn(k, o, o), torch.randn(k, 1, o))
    assert y2.shape == (k, B, o)

    z = y.permute(1, 0, 2)  # (B,k,o)
    logits = torch.randn(B, k)
    pooled = R.attention_pool(z, logits)
    a = torch.softmax(logits, -1)
    ref = (z * a.unsqueeze(-1)).sum(1)
    assert torch.allclose(pooled, ref, atol=1e-6)


def test_dw_arena_split_math():
    """S/chunk selection for the phase arena: uniform 64-row-aligned batch
    splits, ~8 target splits, exact cover of B rows."""
    from distributed_sac_amd.algo.sac import SACEngine
    for B in (16, 64, 192, 256, 1024, 1280, 2560, 1000):
        chunk = ((B + 7) // 8 + 63) // 64 * 64
        S = (B + chunk - 1) // chunk
        assert chunk % 64 == 0
        assert S * chunk >= B > (S - 1) * chunk
        assert S <= 8
