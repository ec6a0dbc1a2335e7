"""Critic (Q) networks — reference-compatible structure and math.

- :class:`Critic` — MT-style twin-Q module with ``Q_function_1`` /
  ``Q_function_2`` (MT10_Distributed_MTSAC/src/model.py:120-196).
- :class:`LLCritic` — LunarLander-style single-Q module (``first_layer`` +
  ``layer_module``; LunarLander_Distributed_SAC/src/model.py:92-142); the
  LL learner instantiates two of these plus two targets.
"""

from __future__ import annotations

from typing import List, Optional

import torch
import torch.nn as nn

from ..ops import functional as Fops
from .mlp import build_mlp, weights_init


class Critic(nn.Module):
    """Twin Q on [mtobs || action]; keys ``Q_function_{1,2}.{0,2,4,6}.*``."""

    def __init__(self, state_dim: int, action_dim: int,
                 hidden_dims: List[int], num_tasks: int = 0):
        super().__init__()
        self.state_dim = state_dim
        self.num_tasks = num_tasks
        self.mtobs_dim = state_dim + num_tasks
        self.action_dim = action_dim
        in_dim = self.mtobs_dim + action_dim
        self.Q_function_1 = build_mlp(in_dim, 1, hidden_dims)
        self.Q_function_2 = build_mlp(in_dim, 1, hidden_dims)
        self.Q_function_1.apply(weights_init)
        self.Q_function_2.apply(weights_init)

    def forward(self, mtobss: torch.Tensor, action: torch.Tensor):
        x = torch.cat([mtobss, action], dim=-1)
        return self.Q_function_1(x), self.Q_function_2(x)

    def cal_loss(self, mtobss, action, td_target_values,
                 use_weighted_loss: bool = False,
                 alphas: Optional[torch.Tensor] = None,
                 degenerate: bool = False):
        """Twin MSE vs TD target, optionally task-weighted (reference
        MT10…MTSAC/src/model.py:157-196).

        DELIBERATE DEVIATION (docs/PARITY.md "weighted loss"): the
        reference multiplies weights (B,) by loss (B,1), broadcasting to
        (B,B), so its mean equals mean(loss)/B and the weights cancel —
        it effectively trains UNWEIGHTED at 1/B scale.  Default here is
        TRUE per-sample weighting (``w.unsqueeze(-1) * loss``); pass
        ``degenerate=True`` (cfg ``weighted_loss_mode="reference"``) to
        reproduce the reference's actual numerics for strict parity."""
        q1, q2 = self.forward(mtobss, action)
        l1 = (td_target_values - q1) ** 2
        l2 = (td_target_values - q2) ** 2
        if use_weighted_loss and alphas is not None:
            w = Fops.task_weights(mtobss[:, -self.num_tasks:], alphas)
            if degenerate:  # reference (B,)*(B,1)->(B,B) broadcast
                l1, l2 = w * l1, w * l2
            else:
                l1, l2 = w.unsqueeze(-1) * l1, w.unsqueeze(-1) * l2
        return l1.mean(), l2.mean()


class LLCritic(nn.Module):
    """Single-Q MLP on [state || action] — keys ``first_layer.*``,
    ``layer_module.{0,1}.*`` (LunarLander…/src/model.py:92-142)."""

    def __init__(self, state_dim: int, action_dim: int,
                 hidden_dim: List[int] = (256, 256)):
        super().__init__()
        self.state_dim = state_dim
        self.action_dim = action_dim
        dims = list(hidden_dim) + [1]
        self.first_layer = nn.Linear(state_dim + action_dim, hidden_dim[0])
        self.layer_module = nn.ModuleList(
            [nn.Linear(i, o) for i, o in zip(dims[:-1], dims[1:])])
        self.apply(weights_init)

    def forward(self, state: torch.Tensor, action: torch.Tensor):
        x = torch.cat([state, action], dim=-1)
        ws = [self.first_layer.weight] + [m.weight for m in self.layer_module]
        bs = [self.first_layer.bias] + [m.bias for m in self.layer_module]
        return Fops.mlp_forward(x, ws, bs)

    def cal_loss(self, state, action, td_target_values):
        q = self.forward(state, action)
        return torch.nn.functional.mse_loss(q, td_target_values)
