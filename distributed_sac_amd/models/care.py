"""CARE actor/critic — state-encoder-conditioned policy and twin Q.

Reference: MT10_Distributed_CARE/src/model.py (Actor :11-159, Critic
:163-269).  Attribute/state_dict naming matches (``state_encoder``,
``mu_log_std_layer``, ``Q_function_1/2``) so CARE checkpoints round-trip.
"""

from __future__ import annotations

from typing import Optional

import torch
import torch.nn as nn

from ..ops import functional as Fops
from .mlp import build_mlp, weights_init
from .state_encoder import stateEncoder


class CAREActor(nn.Module):
    def __init__(self, actor_cfg: dict, encoder_cfg: dict,
                 use_modified_care: bool):
        super().__init__()
        self.state_dim = int(actor_cfg["state_dim"])
        self.num_tasks = encoder_cfg["num_tasks"]
        self.mtobs_dim = self.state_dim + self.num_tasks
        self.policy_input_dim = (encoder_cfg["output_dim_contextEnc"]
                                 + encoder_cfg["output_dim_mixtureEnc"])
        self.action_dim = int(actor_cfg["action_dim"])
        self.action_bound = actor_cfg["action_bound"]
        self.k = (self.action_bound[1] - self.action_bound[0]) / 2
        self.use_modified_care = use_modified_care
        self.state_encoder = stateEncoder(encoder_cfg, use_modified_care)
        self.mu_log_std_layer = build_mlp(self.policy_input_dim,
                                          2 * self.action_dim,
                                          actor_cfg["actor_hidden_dim"])
        self.mu_log_std_layer.apply(weights_init)

    def forward(self, mtobss, z_context, detach_z_encs: bool = False):
        """Returns (mu, log_std_raw) — clamp+exp live in the fused
        sampling op (reference model.py:51-66 returns (mu, std))."""
        enc = self.state_encoder(z_context=z_context, mtobss=mtobss,
                                 detach_z_encs=detach_z_encs)
        x = self.mu_log_std_layer(enc)
        return x[:, : self.action_dim], x[:, self.action_dim:]

    def get_action_log_prob_log_std(self, mtobss, z_context,
                                    detach_z_encs: bool = False,
                                    eps: Optional[torch.Tensor] = None):
        mu, log_std_raw = self.forward(mtobss, z_context, detach_z_encs)
        if eps is None:
            eps = torch.randn_like(mu)
        return Fops.squashed_gaussian(mu, log_std_raw, eps, self.k)

    @torch.no_grad()
    def get_action(self, mtobss, z_context, detach_z_encs: bool = False,
                   stochastic: bool = True):
        mu, log_std_raw = self.forward(mtobss, z_context, detach_z_encs)
        if not stochastic:
            return torch.tanh(mu) * self.k  # reference model.py:114
        eps = torch.randn_like(mu)
        a, _, _ = Fops.squashed_gaussian(mu, log_std_raw, eps, self.k)
        return a

    def cal_loss(self, log_probs, Q_min, alpha, use_weighted_loss=False,
                 mtobss=None, num_tasks=None, alphas=None,
                 degenerate=False):
        # degenerate=True reproduces the reference's (B,)x(B,1)->(B,B)
        # broadcast where the weights cancel (docs/PARITY.md)
        loss = -(Q_min - alpha * log_probs)
        if use_weighted_loss and num_tasks and alphas is not None \
                and mtobss is not None:
            w = Fops.task_weights(mtobss[:, -num_tasks:], alphas)
            loss = (w * loss) if degenerate else w.unsqueeze(-1) * loss
        return loss.mean()


class CARECritic(nn.Module):
    def __init__(self, critic_cfg: dict, encoder_cfg: dict,
                 use_modified_care: bool):
        super().__init__()
        self.state_dim = int(critic_cfg["state_dim"])
        self.num_tasks = encoder_cfg["num_tasks"]
        self.mtobs_dim = self.state_dim + self.num_tasks
        self.policy_input_dim = (encoder_cfg["output_dim_contextEnc"]
                                 + encoder_cfg["output_dim_mixtureEnc"])
        self.action_dim = int(critic_cfg["action_dim"])
        self.use_modified_care = use_modified_care
        self.state_encoder = stateEncoder(encoder_cfg, use_modified_care)
        in_dim = self.policy_input_dim + self.action_dim
        self.Q_function_1 = build_mlp(in_dim, 1, critic_cfg["critic_hidden_dim"])
        self.Q_function_2 = build_mlp(in_dim, 1, critic_cfg["critic_hidden_dim"])
        self.Q_function_1.apply(weights_init)
        self.Q_function_2.apply(weights_init)

    def encode(self, mtobss, z_context, detach_z_encs: bool = False):
        return self.state_encoder(z_context=z_context, mtobss=mtobss,
                                  detach_z_encs=detach_z_encs)

    def forward(self, mtobss, z_context, action, detach_z_encs: bool = False):
        enc = self.encode(mtobss, z_context, detach_z_encs)
        x = torch.cat([enc, action], dim=-1)
        return self.Q_function_1(x), self.Q_function_2(x)

    def cal_loss(self, mtobss, z_context, action, td_target_values,
                 detach_z_encs=False, use_weighted_loss=False,
                 num_tasks=None, alphas=None, degenerate=False):
        q1, q2 = self.forward(mtobss, z_context, action, detach_z_encs)
        l1 = (td_target_values - q1) ** 2
        l2 = (td_target_values - q2) ** 2
        if use_weighted_loss and num_tasks and alphas is not None:
            w = Fops.task_weights(mtobss[:, -num_tasks:], alphas)
            if not degenerate:  # corrected default (docs/PARITY.md)
                w = w.unsqueeze(-1)
            l1, l2 = w * l1, w * l2
        return l1.mean(), l2.mean()
