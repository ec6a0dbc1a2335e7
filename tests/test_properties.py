"""Property-based numerics tests (hypothesis) — the fused-op math holds
across random shapes/values, not just the shipped configs.

Oracle = closed-form reference formulas, independently
restated here; the ops under test are ops/torch_ref.py (the CPU path and
the spec the HIP kernels are unit-tested against on the GPU)."""

import math

import torch
from hypothesis import given, settings, strategies as st

from distributed_sac_amd.ops import torch_ref as R

dims = st.integers(min_value=1, max_value=7)
batches = st.integers(min_value=1, max_value=33)


@settings(max_examples=40, deadline=None)
@given(B=batches, A=dims, k=st.floats(0.5, 3.0), seed=st.integers(0, 10**6))
def test_squashed_gaussian_matches_formula(B, A, k, seed):
    g = torch.Generator().manual_seed(seed)
    mu = torch.randn(B, A, generator=g)
    log_std = torch.randn(B, A, generator=g) * 8  # exercises the clamp
    eps = torch.randn(B, A, generator=g)
    a, lp, ls = R.squashed_gaussian(mu, log_std, eps, k)
    lsc = log_std.clamp(-20, 2)
    u = mu + lsc.exp() * eps
    assert torch.allclose(a, k * torch.tanh(u), atol=1e-6)
    # reference log-prob: logN(u) - log(k(1 - tanh(u)^2 + 1e-6)), summed
    logN = (-0.5 * eps.pow(2) - lsc - 0.5 * math.log(2 * math.pi))
    corr = torch.log(k * (1 - torch.tanh(u).pow(2) + 1e-6))
    assert torch.allclose(lp, (logN - corr).sum(-1, keepdim=True),
                          atol=1e-5, rtol=1e-5)
    assert torch.equal(ls, lsc)
    assert torch.isfinite(lp).all()


@settings(max_examples=40, deadline=None)
@given(B=batches, T=st.integers(1, 12), gamma=st.floats(0.0, 1.0),
       scale=st.floats(0.01, 10.0), seed=st.integers(0, 10**6))
def test_td_target_and_task_weights(B, T, gamma, scale, seed):
    g = torch.Generator().manual_seed(seed)
    r = torch.randn(B, 1, generator=g)
    d = (torch.rand(B, 1, generator=g) < 0.3).float()
    q1 = torch.randn(B, 1, generator=g)
    q2 = torch.randn(B, 1, generator=g)
    lp = torch.randn(B, 1, generator=g)
    alpha = torch.rand(B, 1, generator=g)
    y = R.td_target(r, d, q1, q2, lp, alpha, gamma, scale)
    want = scale * r + gamma * (1 - d) * (torch.min(q1, q2) - alpha * lp)
    assert torch.allclose(y, want, atol=1e-6)
    # done rows depend only on the reward
    assert torch.allclose(y[d.bool().squeeze(-1)],
                          (scale * r)[d.bool().squeeze(-1)])

    oh = torch.nn.functional.one_hot(
        torch.randint(0, T, (B,), generator=g), T).float()
    alphas = torch.rand(T, generator=g) + 0.01
    w = R.task_weights(oh, alphas)
    assert abs(float(w.sum()) - 1.0) < 1e-5
    assert (w > 0).all()
    # same-task samples share a weight
    ti = oh.argmax(1)
    for t in range(T):
        m = ti == t
        if int(m.sum()) > 1:
            assert torch.allclose(w[m], w[m][0].expand(int(m.sum())))


@settings(max_examples=30, deadline=None)
@given(B=batches, T=st.integers(1, 12), seed=st.integers(0, 10**6))
def test_gather_log_alpha_and_entropy(B, T, seed):
    g = torch.Generator().manual_seed(seed)
    oh = torch.nn.functional.one_hot(
        torch.randint(0, T, (B,), generator=g), T).float()
    la = torch.randn(T, generator=g)
    out = R.gather_log_alpha(oh, la)
    assert torch.allclose(out.squeeze(-1), la[oh.argmax(1)], atol=1e-6)

    ls = torch.randn(B, max(1, T % 5 + 1), generator=g)
    ent = R.entropy_from_log_std(ls)
    d = ls.shape[1]
    want = (0.5 * d * (1 + math.log(2 * math.pi)) + ls.sum(-1)).mean()
    assert torch.allclose(ent, want, atol=1e-5)


@settings(max_examples=25, deadline=None)
@given(B=batches, k=st.integers(1, 8), D=st.integers(1, 16),
       seed=st.integers(0, 10**6))
def test_attention_pool_properties(B, k, D, seed):
    g = torch.Generator().manual_seed(seed)
    z = torch.randn(B, k, D, generator=g)
    logits = torch.randn(B, k, generator=g)
    out = R.attention_pool(z, logits)
    # convex combination: inside the per-dim min/max envelope
    assert (out <= z.max(dim=1).values + 1e-5).all()
    assert (out >= z.min(dim=1).values - 1e-5).all()
    # uniform logits -> plain mean; one dominant logit -> that encoder
    assert torch.allclose(R.attention_pool(z, torch.zeros(B, k)),
                          z.mean(1), atol=1e-5)
    hot = torch.full((B, k), -1e9)
    hot[:, 0] = 0.0
    assert torch.allclose(R.attention_pool(z, hot), z[:, 0], atol=1e-5)


@settings(max_examples=20, deadline=None)
@given(n=st.integers(1, 2000), tau=st.floats(0.0, 1.0),
       seed=st.integers(0, 10**6))
def test_polyak_endpoints_and_linearity(n, tau, seed):
    g = torch.Generator().manual_seed(seed)
    t = torch.randn(n, generator=g)
    s = torch.randn(n, generator=g)
    t0 = t.clone()
    R.polyak_([t], [s], tau)
    assert torch.allclose(t, (1 - tau) * t0 + tau * s, atol=1e-6)
