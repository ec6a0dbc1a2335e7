"""Checkpoint IO — reference ``saved_models/.../checkpoint_<it>.tar`` format.

Schema per variant (reference learner.save_checkpoint):
- LL/VSAC (LunarLander…/src/learner.py:144-163): episode/update counter,
  total_step, local_critic_1/2, target_critic_1/2, actor, critic_optimizer,
  actor_optimizer, log_alpha(+optimizer), alpha.
- MTSAC (MT10_Distributed_MTSAC/src/learner.py:157-174): local_critic,
  target_critic instead of the four.
- CARE adds context_encoder + its optimizer
  (MT10_Distributed_CARE/src/learner.py:178-198).

The reference's learner-side resume is broken (``self.actor.optimizer``
AttributeError); ours works.
"""

from __future__ import annotations

import os
from typing import Dict, Optional

import torch


def save_checkpoint(engine, save_dir: str, update_iteration: Optional[int] = None,
                    prefix: str = "checkpoint") -> str:
    os.makedirs(save_dir, exist_ok=True)
    it = engine.update_iteration if update_iteration is None else update_iteration
    state = engine.checkpoint_state()
    state["update_iteration"] = it
    if "episode_idx" in state:  # LL variant counter name
        state["episode_idx"] = it
    path = os.path.join(save_dir, f"{prefix}_{it}.tar")
    torch.save(state, path)
    return path


def load_checkpoint(path: str, map_location="cpu") -> Dict:
    return torch.load(path, map_location=map_location, weights_only=False)


def load_into_engine(engine, path: str) -> None:
    engine.load_checkpoint_state(load_checkpoint(path, engine.device))


def load_actor_for_eval(actor: torch.nn.Module, path: str) -> int:
    """Player-side eval load (reference player.load_model,
    player.py:67-76): reads only the actor state_dict from a .tar."""
    ckpt = load_checkpoint(path)
    actor.load_state_dict(ckpt["actor"])
    return int(ckpt.get("update_iteration", 0))
