"""SACEngine equivalence tests: the engine's update (flat params + fused
Adam + fused ops) must reproduce a straight re-implementation of the
reference learner math (LunarLander…/src/learner.py:203-239 /
MT10…MTSAC/src/learner.py:253-325) bit-for-bit at fp32 tolerance, given
identical weights, batch, and eps draws."""

import copy

import pytest
import torch
import torch.nn as nn
from torch.distributions import Normal

from distributed_sac_amd.algo import SACEngine
from distributed_sac_amd.config import SACConfig


def small_cfg(variant: str) -> SACConfig:
    c = SACConfig()
    c.variant = variant
    c.device = "cpu"
    if variant in ("sac", "vsac"):
        c.state_dim, c.action_dim = 8, 2
        c.actor_hidden_dim = c.critic_hidden_dim = [32, 32]
    else:
        c.state_dim, c.action_dim = 12, 3
        c.num_tasks = 4
        c.actor_hidden_dim = c.critic_hidden_dim = [32, 32]
        c.use_weighted_loss = variant == "mtsac"
    c.batch_size = 16
    return c


def make_batch(cfg: SACConfig, B: int = 16, seed: int = 0):
    g = torch.Generator().manual_seed(seed)
    D = cfg.mtobs_dim
    states = torch.randn(B, D, generator=g)
    next_states = torch.randn(B, D, generator=g)
    if cfg.variant in ("mtsac", "care"):
        # overwrite suffix with valid one-hots
        idx = torch.randint(0, cfg.num_tasks, (B,), generator=g)
        oh = torch.nn.functional.one_hot(idx, cfg.num_tasks).float()
        states[:, -cfg.num_tasks:] = oh
        next_states[:, -cfg.num_tasks:] = oh
    return {
        "states": states,
        "actions": torch.rand(B, cfg.action_dim, generator=g) * 2 - 1,
        "rewards": torch.randn(B, 1, generator=g),
        "next_states": next_states,
        "dones": (torch.rand(B, 1, generator=g) < 0.2).float(),
    }


def ref_sample(actor_mu_std, states, eps, k):
    mu, std = actor_mu_std(states)
    u = mu + std * eps
    action = k * torch.tanh(u)
    lp = (Normal(mu, std).log_prob(u)
          - torch.log(k * (1 - (action / k) ** 2 + 1e-6))).sum(-1, keepdim=True)
    return action, lp, torch.log(std)


@pytest.mark.parametrize("variant", ["sac", "mtsac"])
def test_update_matches_reference_math(variant):
    torch.manual_seed(0)
    cfg = small_cfg(variant)
    engine = SACEngine(cfg, "cpu")
    B = cfg.batch_size

    # reference-side copies of all modules (plain torch, torch.optim.Adam)
    ref = copy.deepcopy({"actor": engine.actor.state_dict()})
    actor_r = copy.deepcopy(engine.actor)
    if variant == "sac":
        c1 = copy.deepcopy(engine.local_critic_1)
        c2 = copy.deepcopy(engine.local_critic_2)
        t1 = copy.deepcopy(engine.target_critic_1)
        t2 = copy.deepcopy(engine.target_critic_2)
        critic_params = list(c1.parameters()) + list(c2.parameters())
    else:
        cr = copy.deepcopy(engine.local_critic)
        tr = copy.deepcopy(engine.target_critic)
        critic_params = list(cr.parameters())
    log_alpha_r = nn.Parameter(engine.log_alpha.detach().clone())
    opt_a = torch.optim.Adam(actor_r.parameters(), lr=cfg.lr_actor)
    opt_c = torch.optim.Adam(critic_params, lr=cfg.lr_critic)
    opt_al = torch.optim.Adam([log_alpha_r], lr=cfg.lr_actor)
    H_bar = torch.tensor([-float(cfg.action_dim)])
    k = engine.actor.k

    for step in range(3):
        batch = make_batch(cfg, B, seed=step)
        eps1 = torch.randn(B, cfg.action_dim)
        eps2 = torch.randn(B, cfg.action_dim)
        engine._eps_queue = [eps1.clone(), eps2.clone()]
        engine.update({k2: v.clone() for k2, v in batch.items()})

        # ---- reference update ----
        states, actions = batch["states"], batch["actions"]
        rewards, next_states, dones = (batch["rewards"], batch["next_states"],
                                       batch["dones"])
        if variant == "mtsac":
            one_hots = states[:, -cfg.num_tasks:]
            alpha = (one_hots @ log_alpha_r.unsqueeze(0).t()).exp().detach()
        else:
            alpha = log_alpha_r.exp().detach()
        opt_a.zero_grad(); opt_c.zero_grad(); opt_al.zero_grad()
        with torch.no_grad():
            na, nlp, _ = ref_sample(actor_r.mu_std, next_states, eps1, k)
            if variant == "sac":
                q1t, q2t = t1(next_states, na), t2(next_states, na)
            else:
                q1t, q2t = tr(next_states, na)
            y = cfg.reward_scale * rewards + cfg.gamma * (1 - dones) * (
                torch.min(q1t, q2t) - alpha * nlp)
        if variant == "sac":
            q_loss = (torch.nn.functional.mse_loss(c1(states, actions), y)
                      + torch.nn.functional.mse_loss(c2(states, actions), y))
        else:
            qa, qb = cr(states, actions)
            l1, l2 = (y - qa) ** 2, (y - qb) ** 2
            alphas_d = log_alpha_r.exp().detach()
            ti = torch.argmax(one_hots, dim=1)
            w = torch.softmax(-alphas_d, 0)[ti].detach()
            w = (w / w.sum()).unsqueeze(-1)
            q_loss = (w * l1).mean() + (w * l2).mean()
        q_loss.backward()
        opt_c.step()

        sa, lp, log_stds = ref_sample(actor_r.mu_std, states, eps2, k)
        if variant == "sac":
            qmin = torch.min(c1(states, sa), c2(states, sa))
            policy_loss = -(qmin - alpha * lp).mean()
        else:
            qa, qb = cr(states, sa)
            qmin = torch.min(qa, qb)
            pl = -(qmin - alpha * lp)
            policy_loss = (w * pl).mean()
        policy_loss.backward()
        opt_a.step()

        if variant == "mtsac":
            la = one_hots @ log_alpha_r.unsqueeze(0).t()
            loss_la = -(la * (lp.detach() + H_bar)).mean()
        else:
            loss_la = -(log_alpha_r * (lp.detach() + H_bar)).mean()
        loss_la.backward()
        opt_al.step()

        with torch.no_grad():
            if variant == "sac":
                for tm, lm in ((t1, c1), (t2, c2)):
                    for tp, sp in zip(tm.parameters(), lm.parameters()):
                        tp.copy_(cfg.tau * sp + (1 - cfg.tau) * tp)
            else:
                for tp, sp in zip(tr.parameters(), cr.parameters()):
                    tp.copy_(cfg.tau * sp + (1 - cfg.tau) * tp)

    # ---- compare all parameters ----
    for n, p in engine.actor.named_parameters():
        rp = dict(actor_r.named_parameters())[n]
        assert torch.allclose(p, rp, atol=2e-6), f"actor param {n} diverged"
    if variant == "sac":
        pairs = [(engine.local_critic_1, c1), (engine.local_critic_2, c2),
                 (engine.target_critic_1, t1), (engine.target_critic_2, t2)]
    else:
        pairs = [(engine.local_critic, cr), (engine.target_critic, tr)]
    for em, rm in pairs:
        for (n, p), (_, rp) in zip(em.named_parameters(), rm.named_parameters()):
            assert torch.allclose(p, rp, atol=2e-6), f"critic param {n} diverged"
    assert torch.allclose(engine.log_alpha, log_alpha_r, atol=2e-6)


def test_update_runs_all_variants():
    for variant in ("sac", "vsac", "mtsac"):
        cfg = small_cfg(variant)
        engine = SACEngine(cfg, "cpu")
        m = engine.update(make_batch(cfg))
        for key in ("critic_loss", "actor_loss", "alpha_loss", "entropy"):
            assert key in m and m[key] == m[key]  # finite / not NaN


def test_targets_initialized_to_critics():
    cfg = small_cfg("mtsac")
    engine = SACEngine(cfg, "cpu")
    for tp, sp in zip(engine.target_critic.parameters(),
                      engine.local_critic.parameters()):
        assert torch.equal(tp, sp)


def test_critic_loss_decreases_on_fixed_batch():
    torch.manual_seed(0)
    cfg = small_cfg("sac")
    engine = SACEngine(cfg, "cpu")
    batch = make_batch(cfg, seed=7)
    losses = [engine.update(batch)["critic_loss"] for _ in range(30)]
    assert losses[-1] < losses[0]
