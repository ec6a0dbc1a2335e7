"""Pure-torch fp32 reference implementations of every hot-path op.

These are (a) the CPU execution path, (b) the numerics oracle the HIP
kernels are unit-tested against, and (c) the executable spec of the
reference's math:

- :func:`mlp_forward`        — reference ``build_mlp`` products
  (MT10_Distributed_MTSAC/src/utils.py:36-57) / LL Actor/Critic forward
  (LunarLander_Distributed_SAC/src/model.py:41-44,120-125).
- :func:`squashed_gaussian`  — tanh-squashed Gaussian rsample + log-prob
  (LunarLander…/src/model.py:45-59; MT10…MTSAC/src/model.py:43-56).
- :func:`td_target`          — Bellman backup
  (LunarLander…/src/learner.py:206-210; MT10…MTSAC/src/learner.py:270-276).
- :func:`task_weights`       — per-task weighted-loss weights
  (MT10_Distributed_MTSAC/src/model.py:99-116,177-196).
- :func:`gather_log_alpha`   — per-sample alpha (MT10…MTSAC/src/learner.py:213-233).
- :func:`entropy_from_log_std` — diagnostic entropy
  (MT10…MTSAC/src/learner.py:311-312).
- :func:`polyak_`            — soft target update (learner.soft_update).

All math is fp32 (the reference never uses mixed precision); the HIP path
may run GEMMs in bf16-in/fp32-accumulate and is tested against these at
bf16-appropriate tolerances.
"""

from __future__ import annotations

import math
from typing import Iterable, Optional, Sequence

import torch
import torch.nn.functional as F

LOG_STD_MIN = -20.0
LOG_STD_MAX = 2.0
_LOG_SQRT_2PI = 0.5 * math.log(2 * math.pi)


def mlp_forward(x: torch.Tensor,
                weights: Sequence[torch.Tensor],
                biases: Sequence[torch.Tensor],
                final_act: Optional[str] = None) -> torch.Tensor:
    """ReLU MLP: hidden layers ReLU-activated, output layer linear.

    Matches reference ``build_mlp`` semantics (hidden Linear+ReLU stack, no
    output activation — MT10…MTSAC/src/utils.py:36-57).
    """
    n = len(weights)
    for i, (w, b) in enumerate(zip(weights, biases)):
        x = F.linear(x, w, b)
        if i < n - 1:
            x = torch.relu(x)
        elif final_act == "relu":
            x = torch.relu(x)
    return x


def squashed_gaussian(mu: torch.Tensor, log_std: torch.Tensor,
                      eps: torch.Tensor, k: float):
    """Tanh-squashed Gaussian sample + summed log-prob.

    Given pre-clamp log_std, applies the reference clamp to [-20, 2]
    (model.forward), then u = mu + std*eps, a = k*tanh(u),
    logp = sum_i [ logN(u_i; mu_i, std_i) - log(k*(1 - tanh(u_i)^2 + 1e-6)) ]
    (reference model.get_action_log_prob…, LunarLander…/src/model.py:51-59).

    Returns (action, log_prob[B,1], log_std_clamped).
    """
    log_std = torch.clamp(log_std, LOG_STD_MIN, LOG_STD_MAX)
    std = torch.exp(log_std)
    u = mu + std * eps
    t = torch.tanh(u)
    action = k * t
    gaussian_log_prob = -0.5 * eps.pow(2) - log_std - _LOG_SQRT_2PI
    log_prob = gaussian_log_prob - torch.log(k * (1 - t.pow(2) + 1e-6))
    return action, log_prob.sum(dim=-1, keepdim=True), log_std


def td_target(rewards: torch.Tensor, dones: torch.Tensor,
              q1_target: torch.Tensor, q2_target: torch.Tensor,
              next_log_probs: torch.Tensor, alpha: torch.Tensor,
              gamma: float, reward_scale: float) -> torch.Tensor:
    """y = scale*r + gamma*(1-d)*(min(Q1t,Q2t) - alpha*logp') — reference
    learner.update_SAC (MT10…MTSAC/src/learner.py:270-276)."""
    return reward_scale * rewards + gamma * (1 - dones) * (
        torch.min(q1_target, q2_target) - alpha * next_log_probs)


def task_weights(one_hots: torch.Tensor, alphas: torch.Tensor) -> torch.Tensor:
    """Per-sample weights for the weighted multi-task loss.

    w_i = softmax(-alpha)[task(i)], renormalized to sum to 1 over the batch
    (reference MT10_Distributed_MTSAC/src/model.py:99-116).  ``alphas`` is
    exp(log_alpha).detach() of shape (num_tasks,).
    """
    task_indices = torch.argmax(one_hots, dim=1)
    w = F.softmax(-alphas, dim=0)[task_indices].detach()
    return w / w.sum()


def gather_log_alpha(one_hots: torch.Tensor, log_alpha: torch.Tensor) -> torch.Tensor:
    """(B, num_tasks) @ (num_tasks, 1) -> (B, 1) per-sample log_alpha
    (reference MT10…MTSAC/src/learner.py:213-233)."""
    return torch.matmul(one_hots, log_alpha.unsqueeze(0).t())


def entropy_from_log_std(log_std: torch.Tensor) -> torch.Tensor:
    """Mean analytic Gaussian entropy: 0.5*d*(1+ln 2pi) + sum(log_std)
    (reference MT10…MTSAC/src/learner.py:311-312)."""
    d = log_std.shape[1]
    return (0.5 * d * (1.0 + math.log(2 * math.pi))
            + log_std.sum(dim=-1)).mean()


@torch.no_grad()
def polyak_(target_params: Iterable[torch.Tensor],
            source_params: Iterable[torch.Tensor], tau: float) -> None:
    """theta_target <- tau*theta + (1-tau)*theta_target (reference
    learner.soft_update; tau=1.0 is the hard copy)."""
    for tp, sp in zip(target_params, source_params):
        tp.mul_(1.0 - tau).add_(sp, alpha=tau)


# ---------------------------------------------------------------------------
# CARE mixture-of-encoders + attention pool (reference
# MT10_Distributed_CARE/src/state_encoder.py:85-94,146-174).
# ---------------------------------------------------------------------------

def batched_linear(x: torch.Tensor, w: torch.Tensor, b: torch.Tensor) -> torch.Tensor:
    """k parallel Linear layers: w (k,in,out), b (k,1,out).

    x (B,in) -> (k,B,out), or x (k,B,in) -> (k,B,out) — the reference's
    einsum pair ('kio,bi->kbo' / 'kio,kbi->kbo',
    MT10_Distributed_CARE/src/state_encoder.py:146-174).
    """
    if x.dim() == 2:
        return torch.einsum("kio,bi->kbo", w, x) + b
    return torch.einsum("kio,kbi->kbo", w, x) + b


def attention_pool(z_encs: torch.Tensor, logits: torch.Tensor) -> torch.Tensor:
    """softmax(logits) convex combination of per-encoder embeddings.

    z_encs (B,k,D), logits (B,k) -> (B,D)
    (reference state_encoder.py:85-89).
    """
    alpha = F.softmax(logits, dim=-1)
    return (z_encs * alpha.unsqueeze(-1)).sum(dim=1)
