"""CARE context encoder — frozen task-language embeddings.

Re-implementation of reference MT10_Distributed_CARE/src/context_encoder.py
(6-127) with identical state_dict naming:

- Modified CARE: ``embedding`` = Sequential(frozen nn.Embedding) returning
  the raw 768-d pretrained (RoBERTa) sentence embedding per task (:54-58);
  the projection lives in stateEncoder.mlp_context instead.
- Original CARE: ``embedding`` = Sequential(frozen Embedding, ReLU,
  embedding_header[768->2E->E with ReLUs]), then ``mlp`` (build_mlp E->...)
  (:59-89); the MT1 copy is original-only (MT1…/src/context_encoder.py:10).

Task index comes from argmax over the mtobs one-hot suffix (:91-106).
Embeddings load from the reference metadata JSON format
(cfg/metadata/*_pretrained_embedding.json: task name -> 768-d list).
"""

from __future__ import annotations

import torch
import torch.nn as nn

import os

from ..config import cfg_read
from .mlp import build_mlp, weights_init

_REPO_ROOT = os.path.dirname(os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))))


def _resolve(path: str) -> str:
    if os.path.isabs(path) or os.path.exists(path):
        return path
    alt = os.path.join(_REPO_ROOT, path)
    return alt if os.path.exists(alt) else path


class contextEncoder(nn.Module):
    def __init__(self, encoder_cfg: dict, use_modified_care: bool):
        super().__init__()
        self.hidden_dims = encoder_cfg["hidden_dims_contextEnc"]
        self.embedding_dim = encoder_cfg["embedding_dim_contextEnc"]
        self.output_dim = encoder_cfg["output_dim_contextEnc"]
        self.use_modified_care = use_modified_care

        emb_map = cfg_read(_resolve(encoder_cfg["pretrained_embedding_json_path"]))
        self.task_name_list = cfg_read(_resolve(encoder_cfg["task_name_json_path"]))
        self.num_tasks = len(self.task_name_list)
        table = torch.tensor([emb_map[name] for name in self.task_name_list],
                             dtype=torch.float32)
        frozen = nn.Embedding.from_pretrained(embeddings=table, freeze=True)

        if use_modified_care:
            self.embedding = nn.Sequential(frozen)
        else:
            header = nn.Sequential(
                nn.Linear(table.shape[1], 2 * self.embedding_dim), nn.ReLU(),
                nn.Linear(2 * self.embedding_dim, self.embedding_dim),
                nn.ReLU())
            header.apply(weights_init)
            self.embedding = nn.Sequential(frozen, nn.ReLU(), header)
            self.mlp = build_mlp(self.embedding_dim, self.output_dim,
                                 self.hidden_dims)
            self.mlp.apply(weights_init)

    def mtobss2states_taskIndices(self, mtobss: torch.Tensor):
        one_hots = mtobss[:, -self.num_tasks:]
        assert one_hots.shape[1] == self.num_tasks
        states = mtobss[:, : -self.num_tasks]
        return states, torch.argmax(one_hots, dim=1)

    def forward(self, mtobss: torch.Tensor) -> torch.Tensor:
        _, task_indices = self.mtobss2states_taskIndices(mtobss)
        if self.use_modified_care:
            return self.embedding(task_indices)
        return self.mlp(self.embedding(task_indices))
