"""CARE state encoder — mixture of k feed-forward encoders + attention.

Re-implementation of reference MT10_Distributed_CARE/src/state_encoder.py
(7-221) with identical state_dict naming and math:

- :class:`Linear` — batched k-encoder linear layer holding W (k,in,out) /
  b (k,1,out), applied via the einsum pair ('kio,bi->kbo' /
  'kio,kbi->kbo') (reference :146-174);
- :class:`feedForwardEncoder` — k parallel Linear+ReLU stacks (:186-221);
- :class:`stateEncoder` — attention trunk over z_context.detach() ->
  softmax over k encoders -> convex combination; modified CARE projects
  z_context through ``mlp_context`` (:50-57, 85-94); ``detach_z_encs``
  stops actor-loss gradients into the mixture (:124-126).

Quirk preserved: ``z_enc / alpha.sum(dim=1)`` (a numeric no-op since alpha
is a softmax, reference :89) is kept for bit-parity.
"""

from __future__ import annotations

import torch
import torch.nn as nn

from ..ops import functional as Fops
from .mlp import build_mlp, weights_init


class Linear(nn.Module):
    """Batched linear for the mixture of encoders (reference :135-174)."""

    def __init__(self, num_encoders: int, in_features: int, out_features: int):
        super().__init__()
        self.num_encoders = num_encoders
        self.in_features = in_features
        self.out_features = out_features
        self.W = nn.Parameter(torch.randn(num_encoders, in_features, out_features))
        self.b = nn.Parameter(torch.randn(num_encoders, 1, out_features))

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return Fops.batched_linear(x, self.W, self.b)


class feedForwardEncoder(nn.Module):
    """k parallel MLP encoders -> (B, k, out) (reference :186-221)."""

    def __init__(self, num_encoders: int, input_dim: int, output_dim: int,
                 hidden_dims=(50, 50)):
        super().__init__()
        layers = []
        dims = [input_dim] + list(hidden_dims)
        for i, o in zip(dims[:-1], dims[1:]):
            layers.append(Linear(num_encoders, i, o))
            layers.append(nn.ReLU())
        layers.append(Linear(num_encoders, hidden_dims[-1], output_dim))
        self.mixtureEncoders = nn.Sequential(*layers)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return self.mixtureEncoders(x).transpose(1, 0)


class stateEncoder(nn.Module):
    """Mixture-of-encoders + attention pooling (reference :7-131)."""

    def __init__(self, encoder_cfg: dict, use_modified_care: bool):
        super().__init__()
        self.hidden_dims_mixtureEnc = encoder_cfg["hidden_dims_mixtureEnc"]
        self.num_tasks = encoder_cfg["num_tasks"]
        self.use_modified_care = use_modified_care
        if use_modified_care:
            self.input_dim = encoder_cfg["RoBERTa_embedding_dim"]
        else:
            self.input_dim = int(encoder_cfg["embedding_dim_contextEnc"])
        self.mixture_encoders = feedForwardEncoder(
            num_encoders=encoder_cfg["num_encoders"],
            input_dim=int(encoder_cfg["state_dim"]),
            output_dim=int(encoder_cfg["output_dim_mixtureEnc"]),
            hidden_dims=encoder_cfg["hidden_dims_mixtureEnc"])
        self.trunk = build_mlp(
            input_dim=self.input_dim,
            output_dim=int(encoder_cfg["num_encoders"]),
            hidden_dims=encoder_cfg["hidden_dims_mixtureEnc"])
        self.trunk.apply(weights_init)
        self.softmax = nn.Softmax(dim=-1)
        if use_modified_care:
            self.mlp_context = build_mlp(
                input_dim=self.input_dim,
                output_dim=int(encoder_cfg["output_dim_contextEnc"]),
                hidden_dims=encoder_cfg["hidden_dims_contextEnc"])
            self.mlp_context.apply(weights_init)

    def mtobss2states_taskIndices(self, mtobss: torch.Tensor):
        one_hots = mtobss[:, -self.num_tasks:]
        assert one_hots.shape[1] == self.num_tasks
        states = mtobss[:, : -self.num_tasks]
        return states, torch.argmax(one_hots, dim=1)

    def encode_states(self, z_encs: torch.Tensor, z_context: torch.Tensor):
        """alpha = softmax(trunk(z_context.detach())); convex combination
        (reference :76-98)."""
        alpha = self.trunk(z_context.detach())
        alpha = self.softmax(alpha).unsqueeze(dim=-1)
        z_enc = (z_encs * alpha).sum(dim=1)
        z_enc = z_enc / alpha.sum(dim=1)  # no-op; reference parity (:89)
        if self.use_modified_care:
            z_context = self.mlp_context(z_context)
        return torch.cat([z_context, z_enc], dim=1)

    def attention_alphas(self, z_context: torch.Tensor) -> torch.Tensor:
        """Diagnostic: the k attention weights (plot_utils parity)."""
        return self.softmax(self.trunk(z_context.detach()))

    def forward(self, z_context: torch.Tensor, mtobss: torch.Tensor,
                detach_z_encs: bool = False) -> torch.Tensor:
        states, _ = self.mtobss2states_taskIndices(mtobss)
        z_encs = self.mixture_encoders(states)
        if not self.use_modified_care:
            assert z_context.shape[-1] == z_encs.shape[-1]
        if detach_z_encs:
            z_encs = z_encs.detach()
        return self.encode_states(z_encs, z_context)
