"""Offline CARE analysis — reference plot_utils parity.

Reference (MT10_Distributed_CARE/src/plot_utils/):
- ``cal_z_context.py:15-35`` — load a checkpoint, print per-task z_context
  and attention alphas (note: the reference script imports a stale module
  path and cannot run as-is; this one works);
- ``plot_attention_map_encoder_alphas.py:7-81`` — heatmap of the k
  attention weights per task;
- ``plot_cosine_similarity_map_z_context.py:7-63`` — cosine-similarity
  matrix of the per-task context embeddings.

Works from a CARE checkpoint ``.tar`` + its cfg.
"""

from __future__ import annotations

from typing import Optional

import numpy as np
import torch

from ..checkpoint import load_checkpoint
from ..config import SACConfig


def _load_care(cfg: SACConfig, checkpoint_path: str):
    from ..models.care import CARECritic
    from ..models.context_encoder import contextEncoder
    enc_cfg = dict(cfg.encoder)
    enc_cfg.setdefault("RoBERTa_embedding_dim", 768)
    ckpt = load_checkpoint(checkpoint_path)
    ctx = contextEncoder(enc_cfg, cfg.use_modified_care)
    ctx.load_state_dict(ckpt["context_encoder"])
    critic = CARECritic(
        {"state_dim": cfg.state_dim, "action_dim": cfg.action_dim,
         "critic_hidden_dim": cfg.critic_hidden_dim},
        enc_cfg, cfg.use_modified_care)
    critic.load_state_dict(ckpt["local_critic"])
    ctx.eval()
    critic.eval()
    return ctx, critic


@torch.no_grad()
def cal_z_context(cfg: SACConfig, checkpoint_path: str) -> torch.Tensor:
    """Per-task context embedding matrix (num_tasks, z_dim)."""
    ctx, _ = _load_care(cfg, checkpoint_path)
    T = cfg.num_tasks
    mtobss = torch.zeros(T, cfg.mtobs_dim)
    mtobss[:, -T:] = torch.eye(T)
    return ctx(mtobss)


@torch.no_grad()
def attention_map(cfg: SACConfig, checkpoint_path: str,
                  states: Optional[torch.Tensor] = None) -> np.ndarray:
    """(num_tasks, num_encoders) mean attention alphas of the critic's
    state encoder over (given or zero) states per task."""
    ctx, critic = _load_care(cfg, checkpoint_path)
    T = cfg.num_tasks
    mtobss = torch.zeros(T, cfg.mtobs_dim)
    if states is not None:
        mtobss[:, : cfg.state_dim] = states
    mtobss[:, -T:] = torch.eye(T)
    z = ctx(mtobss)
    return critic.state_encoder.attention_alphas(z).numpy()


@torch.no_grad()
def z_context_cosine_similarity(cfg: SACConfig,
                                checkpoint_path: str) -> np.ndarray:
    z = cal_z_context(cfg, checkpoint_path)
    z = z / z.norm(dim=1, keepdim=True).clamp_min(1e-12)
    return (z @ z.t()).numpy()


def plot_attention_map(cfg: SACConfig, checkpoint_path: str,
                       out_path: str = "attention_map.png") -> str:
    import matplotlib
    matplotlib.use("Agg")
    import matplotlib.pyplot as plt
    amap = attention_map(cfg, checkpoint_path)
    fig, ax = plt.subplots(figsize=(8, 6))
    im = ax.imshow(amap, cmap="viridis", aspect="auto")
    ax.set_xlabel("encoder")
    ax.set_ylabel("task")
    fig.colorbar(im)
    fig.savefig(out_path, bbox_inches="tight")
    plt.close(fig)
    return out_path


def plot_cosine_similarity_map(cfg: SACConfig, checkpoint_path: str,
                               out_path: str = "z_context_cosine.png") -> str:
    import matplotlib
    matplotlib.use("Agg")
    import matplotlib.pyplot as plt
    sim = z_context_cosine_similarity(cfg, checkpoint_path)
    fig, ax = plt.subplots(figsize=(7, 6))
    im = ax.imshow(sim, cmap="coolwarm", vmin=-1, vmax=1)
    ax.set_xlabel("task")
    ax.set_ylabel("task")
    fig.colorbar(im)
    fig.savefig(out_path, bbox_inches="tight")
    plt.close(fig)
    return out_path
