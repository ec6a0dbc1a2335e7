"""Actor (policy) networks — reference-compatible structure and math.

Two families:

- :class:`Actor` — the MT-style actor (MT10_Distributed_MTSAC/src/model.py:
  9-116): one ``mu_log_std_layer`` MLP over (mt)obs producing [mu | log_std],
  log_std clamped to [-20, 2], tanh-squashed Gaussian actions scaled by k.
- :class:`LLActor` — the LunarLander-style actor (LunarLander_Distributed_
  SAC/src/model.py:7-88): ``layer_intermediate`` ModuleList + separate
  ``mu_log_std_layer`` Linear; checkpoint keys match the reference so the
  shipped checkpoint_165000.tar loads.

Sampling runs through the fused squashed-Gaussian op with externally drawn
eps (graph-capturable, deterministic under torch seeds).
"""

from __future__ import annotations

from typing import List, Optional, Tuple

import torch
import torch.nn as nn

from ..ops import functional as Fops
from .mlp import build_mlp, weights_init


class Actor(nn.Module):
    """MT-style actor: mu_log_std_layer over mtobs.

    state_dict keys: ``mu_log_std_layer.{0,2,4,6}.{weight,bias}`` — identical
    to reference build_mlp products (MT10…MTSAC/src/model.py:27-33).
    """

    def __init__(self, state_dim: int, action_dim: int,
                 hidden_dims: List[int], action_bound=( -1.0, 1.0),
                 num_tasks: int = 0):
        super().__init__()
        self.state_dim = state_dim
        self.num_tasks = num_tasks
        self.mtobs_dim = state_dim + num_tasks
        self.action_dim = action_dim
        self.action_bound = list(action_bound)
        self.k = (self.action_bound[1] - self.action_bound[0]) / 2
        self.mu_log_std_layer = build_mlp(self.mtobs_dim, 2 * action_dim, hidden_dims)
        self.mu_log_std_layer.apply(weights_init)

    def forward(self, mtobss: torch.Tensor) -> Tuple[torch.Tensor, torch.Tensor]:
        """Returns (mu, log_std_raw).  NOTE: unlike the reference's (mu, std),
        the clamp+exp runs inside the fused sampling op; use
        :meth:`mu_std` for the reference-identical pair."""
        x = self.mu_log_std_layer(mtobss)
        return x[:, : self.action_dim], x[:, self.action_dim:]

    def mu_std(self, mtobss: torch.Tensor):
        mu, log_std_raw = self.forward(mtobss)
        return mu, torch.exp(torch.clamp(log_std_raw, -20, 2))

    def get_action_log_prob_log_std(self, mtobss: torch.Tensor,
                                    eps: Optional[torch.Tensor] = None):
        """(action, log_prob[B,1], log_std) — reference
        get_action_log_prob_log_std (MT10…MTSAC/src/model.py:43-56)."""
        mu, log_std_raw = self.forward(mtobss)
        if eps is None:
            eps = torch.randn_like(mu)
        return Fops.squashed_gaussian(mu, log_std_raw, eps, self.k)

    def get_action_log_prob(self, mtobss: torch.Tensor,
                            eps: Optional[torch.Tensor] = None):
        a, lp, _ = self.get_action_log_prob_log_std(mtobss, eps)
        return a, lp

    @torch.no_grad()
    def get_action(self, mtobss: torch.Tensor, stochastic: bool = True):
        """Env-side action selection (reference model.get_action,
        MT10…MTSAC/src/model.py:58-71: deterministic = k*tanh(mu))."""
        mu, log_std_raw = self.forward(mtobss)
        if not stochastic:
            return self.k * torch.tanh(mu)
        eps = torch.randn_like(mu)
        a, _, _ = Fops.squashed_gaussian(mu, log_std_raw, eps, self.k)
        return a

    def cal_loss(self, log_probs, Q_min, alpha, use_weighted_loss=False,
                 mtobss=None, alphas=None, degenerate=False):
        """Policy loss -(Qmin - alpha*logpi), optionally task-weighted
        (reference MT10…MTSAC/src/model.py:80-116).

        DELIBERATE DEVIATION (docs/PARITY.md "weighted loss"): the
        reference's (B,)x(B,1) broadcast makes the weights cancel
        (= mean(loss)/B); default here is true per-sample weighting,
        ``degenerate=True`` reproduces the reference numerics."""
        loss = -(Q_min - alpha * log_probs)
        if use_weighted_loss and alphas is not None and mtobss is not None:
            w = Fops.task_weights(mtobss[:, -self.num_tasks:], alphas)
            loss = (w * loss) if degenerate else w.unsqueeze(-1) * loss
        return loss.mean()


class LLActor(nn.Module):
    """LunarLander-style actor — checkpoint-key compatible with
    LunarLander_Distributed_SAC/src/model.py:7-88.

    Quirk preserved: the reference's deterministic eval action is ``mu * k``
    WITHOUT tanh (model.py:63,80) — we keep that for checkpoint replay
    parity, gated by ``faithful_eval``.
    """

    def __init__(self, state_dim: int, action_dim: int,
                 hidden_dim: List[int] = (256, 256),
                 action_bound=(-1.0, 1.0), faithful_eval: bool = True):
        super().__init__()
        self.state_dim = state_dim
        self.action_dim = action_dim
        self.action_bound = list(action_bound)
        self.k = (self.action_bound[1] - self.action_bound[0]) / 2
        self.faithful_eval = faithful_eval
        dims = [state_dim] + list(hidden_dim)
        self.layer_intermediate = nn.ModuleList(
            [nn.Linear(i, o) for i, o in zip(dims[:-1], dims[1:])])
        self.mu_log_std_layer = nn.Linear(dims[-1], 2 * action_dim)
        self.apply(weights_init)

    def _trunk_weights(self):
        ws = [m.weight for m in self.layer_intermediate] + [self.mu_log_std_layer.weight]
        bs = [m.bias for m in self.layer_intermediate] + [self.mu_log_std_layer.bias]
        return ws, bs

    def forward(self, state: torch.Tensor):
        ws, bs = self._trunk_weights()
        x = Fops.mlp_forward(state, ws, bs)
        return x[:, : self.action_dim], x[:, self.action_dim:]

    def mu_std(self, state: torch.Tensor):
        mu, log_std_raw = self.forward(state)
        return mu, torch.exp(torch.clamp(log_std_raw, -20, 2))

    def get_action_log_prob(self, state: torch.Tensor,
                            eps: Optional[torch.Tensor] = None):
        mu, log_std_raw = self.forward(state)
        if eps is None:
            eps = torch.randn_like(mu)
        a, lp, _ = Fops.squashed_gaussian(mu, log_std_raw, eps, self.k)
        return a, lp

    @torch.no_grad()
    def get_action(self, state: torch.Tensor, stochastic: bool = True):
        mu, log_std_raw = self.forward(state)
        if not stochastic:
            # reference LunarLander model.py:80: action = mu * k (no tanh)
            return mu * self.k if self.faithful_eval else self.k * torch.tanh(mu)
        eps = torch.randn_like(mu)
        a, _, _ = Fops.squashed_gaussian(mu, log_std_raw, eps, self.k)
        return a

    def cal_loss(self, log_probs, Q_min, alpha):
        """-mean(Qmin - alpha*logpi) (LunarLander…/src/model.py:84-88)."""
        return -(Q_min - alpha * log_probs).mean()
