"""Model structure/compatibility tests: state_dict keys must match the
reference naming so shipped checkpoints load."""

import torch
import torch.nn as nn

from distributed_sac_amd.models import Actor, Critic, LLActor, LLCritic


def test_mt_actor_keys():
    a = Actor(39, 4, [400, 400, 400], num_tasks=10)
    keys = set(a.state_dict().keys())
    # reference build_mlp => Sequential indices 0,2,4 hidden + 6 output
    expected = {f"mu_log_std_layer.{i}.{p}" for i in (0, 2, 4, 6)
                for p in ("weight", "bias")}
    assert keys == expected
    assert a.state_dict()["mu_log_std_layer.0.weight"].shape == (400, 49)
    assert a.state_dict()["mu_log_std_layer.6.weight"].shape == (8, 400)


def test_mt_critic_keys():
    c = Critic(39, 4, [400, 400, 400], num_tasks=10)
    keys = set(c.state_dict().keys())
    expected = {f"Q_function_{q}.{i}.{p}" for q in (1, 2) for i in (0, 2, 4, 6)
                for p in ("weight", "bias")}
    assert keys == expected
    assert c.state_dict()["Q_function_1.0.weight"].shape == (400, 53)


def test_ll_actor_keys_and_forward():
    a = LLActor(8, 2, [256, 256])
    keys = set(a.state_dict().keys())
    expected = {f"layer_intermediate.{i}.{p}" for i in (0, 1)
                for p in ("weight", "bias")} | {
        "mu_log_std_layer.weight", "mu_log_std_layer.bias"}
    assert keys == expected
    x = torch.randn(5, 8)
    mu, ls = a(x)
    assert mu.shape == (5, 2) and ls.shape == (5, 2)
    # manual forward equivalence
    h = torch.relu(a.layer_intermediate[0](x))
    h = torch.relu(a.layer_intermediate[1](h))
    out = a.mu_log_std_layer(h)
    assert torch.allclose(mu, out[:, :2], atol=1e-6)
    assert torch.allclose(ls, out[:, 2:], atol=1e-6)


def test_ll_critic_keys_and_forward():
    c = LLCritic(8, 2, [256, 256])
    keys = set(c.state_dict().keys())
    expected = {"first_layer.weight", "first_layer.bias"} | {
        f"layer_module.{i}.{p}" for i in (0, 1) for p in ("weight", "bias")}
    assert keys == expected
    s, a = torch.randn(7, 8), torch.randn(7, 2)
    q = c(s, a)
    x = torch.cat([s, a], -1)
    h = torch.relu(c.first_layer(x))
    h = torch.relu(c.layer_module[0](h))
    ref = c.layer_module[1](h)
    assert torch.allclose(q, ref, atol=1e-6)


def test_fused_mlp_matches_plain_sequential_module():
    from distributed_sac_amd.models import build_mlp
    torch.manual_seed(0)
    m = build_mlp(10, 3, [32, 32])
    plain = nn.Sequential(*[mod for mod in m])
    x = torch.randn(9, 10)
    y = m(x)
    ref = x
    for mod in plain:
        ref = mod(ref)
    assert torch.allclose(y, ref, atol=1e-6)


def test_actor_action_bounds_and_determinism():
    torch.manual_seed(0)
    a = Actor(39, 4, [64, 64], num_tasks=10, action_bound=[-2.0, 2.0])
    x = torch.randn(11, 49)
    act = a.get_action(x, stochastic=True)
    assert act.abs().max() <= 2.0
    d1 = a.get_action(x, stochastic=False)
    d2 = a.get_action(x, stochastic=False)
    assert torch.equal(d1, d2)
    mu, _ = a(x)
    assert torch.allclose(d1, 2.0 * torch.tanh(mu), atol=1e-6)


def test_ll_actor_faithful_eval_quirk():
    """Reference LL deterministic action = mu*k WITHOUT tanh
    (LunarLander…/src/model.py:80)."""
    torch.manual_seed(1)
    a = LLActor(8, 2, [32, 32], action_bound=[-1.0, 1.0])
    x = torch.randn(5, 8)
    mu, _ = a(x)
    det = a.get_action(x, stochastic=False)
    assert torch.allclose(det, mu * a.k, atol=1e-6)
