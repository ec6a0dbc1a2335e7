"""Fused MLP module with reference-compatible state_dict naming.

``build_mlp`` mirrors the reference factory (MT10_Distributed_MTSAC/src/
utils.py:36-57): hidden Linear+ReLU stack, linear output layer, Xavier-
uniform weights / zero bias.  The module subclasses ``nn.Sequential`` so
state_dict keys are identical (``0.weight``, ``2.weight``, …), but forward
is routed through :func:`distributed_sac_amd.ops.functional.mlp_forward`,
which on GPU runs the whole chain as fused HIP MFMA kernels.
"""

from __future__ import annotations

from typing import List

import torch
import torch.nn as nn

from ..ops import functional as Fops


def weights_init(m: nn.Module) -> None:
    """Xavier-uniform + zero bias (reference utils.weights_init,
    MT10_Distributed_MTSAC/src/utils.py:30-33)."""
    if isinstance(m, nn.Linear):
        nn.init.xavier_uniform_(m.weight, gain=1)
        nn.init.constant_(m.bias, 0)


class FusedMLP(nn.Sequential):
    """Sequential(Linear, ReLU, ..., Linear) with fused execution."""

    def forward(self, x: torch.Tensor) -> torch.Tensor:  # noqa: D102
        ws: List[torch.Tensor] = []
        bs: List[torch.Tensor] = []
        for m in self:
            if isinstance(m, nn.Linear):
                ws.append(m.weight)
                bs.append(m.bias)
        return Fops.mlp_forward(x, ws, bs)


def build_mlp(input_dim: int, output_dim: int, hidden_dims: List[int]) -> FusedMLP:
    """Reference build_mlp layout: no activation on the output layer."""
    layers: List[nn.Module] = []
    dims = [input_dim] + list(hidden_dims)
    for d_in, d_out in zip(dims[:-1], dims[1:]):
        layers.append(nn.Linear(d_in, d_out))
        layers.append(nn.ReLU())
    layers.append(nn.Linear(hidden_dims[-1], output_dim))
    return FusedMLP(*layers)
