"""Player process — asynchronous rollout worker.

Re-implements the reference Player process (LunarLander_Distributed_SAC/
src/player.py:13-153, MT10_Distributed_MTSAC/src/player.py:13-286) without
Ray or Redis:

- transitions leave through a torch.multiprocessing queue in BLOCKS (not
  per-step pickles — reference C1 in SURVEY §2.7);
- weights arrive through the shared-memory ParamSnapshot seqlock: the
  16-byte version word is polled once per collect chunk and the flat
  buffer copied only when update_iteration advanced (keeping reference
  pull_parameters semantics, player.py:75-85, at ~0 cost);
- each worker owns SEVERAL envs (possibly spanning tasks — reference MT10
  task round-robin player.py:247-253) stepped in lockstep with batched
  inference via VecRollout.

Eval mode (reference main.py is_train=False branch): load a checkpoint
actor and run deterministic episodes — :func:`evaluate_checkpoint`.
"""

from __future__ import annotations

import os
import queue as pyqueue
import time
from typing import Callable, Dict, List, Optional

import numpy as np
import torch
import torch.nn as nn

from ..checkpoint import load_actor_for_eval
from ..config import SACConfig
from ..models import Actor, LLActor
from .param_server import ParamSnapshot
from .rollout import VecRollout


def build_actor(cfg: SACConfig, device="cpu") -> nn.Module:
    if cfg.variant in ("sac", "vsac"):
        return LLActor(cfg.state_dim, cfg.action_dim, cfg.actor_hidden_dim,
                       cfg.action_bound).to(device)
    if cfg.variant == "care":
        return CAREPolicy(cfg).to(device)
    return Actor(cfg.state_dim, cfg.action_dim, cfg.actor_hidden_dim,
                 cfg.action_bound, num_tasks=cfg.num_tasks).to(device)


class CAREPolicy(nn.Module):
    """Rollout-side CARE bundle: context encoder + actor with the reference
    Player's action path (MT10_Distributed_CARE/src/player.py:202-209:
    z_context = context_encoder(mtobs); actor.get_action(mtobs, z))."""

    def __init__(self, cfg: SACConfig):
        super().__init__()
        from ..models.care import CAREActor
        from ..models.context_encoder import contextEncoder
        enc_cfg = dict(cfg.encoder)
        enc_cfg.setdefault("RoBERTa_embedding_dim", 768)
        self.context_encoder = contextEncoder(enc_cfg, cfg.use_modified_care)
        self.actor = CAREActor(
            {"state_dim": cfg.state_dim, "action_dim": cfg.action_dim,
             "action_bound": cfg.action_bound,
             "actor_hidden_dim": cfg.actor_hidden_dim},
            enc_cfg, cfg.use_modified_care)
        self.k = self.actor.k

    @torch.no_grad()
    def get_action(self, mtobss: torch.Tensor, stochastic: bool = True):
        z = self.context_encoder(mtobss)
        return self.actor.get_action(mtobss, z, stochastic=stochastic)

    def load_state_dict_from_checkpoint(self, ckpt):
        self.actor.load_state_dict(ckpt["actor"])
        if "context_encoder" in ckpt:
            self.context_encoder.load_state_dict(ckpt["context_encoder"])


def policy_params(policy: nn.Module):
    """Ordered trainable params matching SACEngine/CAREEngine
    publish_params: [actor params | trainable context params]."""
    if isinstance(policy, CAREPolicy):
        return (list(policy.actor.parameters())
                + [p for p in policy.context_encoder.parameters()
                   if p.requires_grad])
    return list(policy.parameters())


@torch.no_grad()
def apply_flat_params(actor: nn.Module, flat: torch.Tensor) -> None:
    """Copy a flat fp32 vector (engine publish_params order) into the
    local policy."""
    torch.nn.utils.vector_to_parameters(flat, policy_params(actor))


def run_player(player_id: int, cfg: SACConfig, env_fn: Callable,
               task_idx_list: List[int], snapshot: ParamSnapshot,
               sample_queue, log_queue=None, stop_event=None,
               chunk_steps: int = 64, seed: int = 0,
               print_period_episodes: int = 0,
               eval_every_episodes: int = 0,
               max_chunks: Optional[int] = None,
               heartbeat: Optional[torch.Tensor] = None,
               ring_name: Optional[str] = None) -> None:
    """Infinite rollout loop (reference Player.run)."""
    # rollout inference is tiny (a handful of envs): one intra-op thread.
    # torch's default pool (~cores/2 PER PROCESS) oversubscribes the host
    # and starves the learner's HIP runtime threads (measured: host-side
    # GPU calls stretched from us to ms with 2 default-pool players).
    torch.set_num_threads(1)
    torch.manual_seed(seed)
    ring = None
    if ring_name is not None:
        from .. import ops
        ring = ops.native().ShmRing.open(ring_name)
    actor = build_actor(cfg)
    actor.eval()
    envs = [env_fn(cfg, t, seed * 7919 + i) for i, t in enumerate(task_idx_list)]
    rollout = VecRollout(cfg, envs, task_idx_list, actor, device="cpu",
                         seed=seed)
    last_iteration = -1
    dropped_blocks = 0
    flat = torch.zeros(snapshot.buf.numel())
    episodes_seen = {t: 0 for t in set(task_idx_list)}
    eval_done_at = {t: 0 for t in set(task_idx_list)}
    n_chunks = 0
    while stop_event is None or not stop_event.is_set():
        # pull parameters when a newer snapshot exists (reference
        # pull_parameters: apply only when update_iteration changed)
        if heartbeat is not None:
            heartbeat[player_id] = time.time()
        it = snapshot.read(flat, last_iteration)
        if it is not None:
            apply_flat_params(actor, flat)
            last_iteration = it
        blocks = rollout.collect(chunk_steps)
        for t, blk in blocks.items():
            if not blk["states"].shape[0]:
                continue
            if ring is not None:
                tensors = (torch.from_numpy(blk["states"]),
                           torch.from_numpy(blk["actions"]),
                           torch.from_numpy(blk["rewards"].reshape(-1, 1)),
                           torch.from_numpy(blk["next_states"]),
                           torch.from_numpy(blk["dones"].reshape(-1, 1)))
                # bounded backoff instead of fire-and-forget: the reference
                # (Redis list) never dropped data; we retry ~100 ms before
                # conceding, and LOG the drop player-side (the shared ring
                # header also counts it for the learner)
                ok = ring.push(t, *tensors)
                attempts = 0
                while not ok and attempts < 20:
                    if stop_event is not None and stop_event.is_set():
                        break
                    time.sleep(0.005)  # learner behind
                    ok = ring.push(t, *tensors)
                    attempts += 1
                if not ok:
                    dropped_blocks += 1
                    if dropped_blocks <= 10 or dropped_blocks % 100 == 0:
                        print(f"[player {player_id}] ring full after "
                              f"{attempts} retries — dropped block "
                              f"(task {t}, {blk['states'].shape[0]} steps, "
                              f"{dropped_blocks} total drops)", flush=True)
            else:
                try:
                    sample_queue.put((player_id, t, blk), timeout=5.0)
                except pyqueue.Full:  # learner stalled: drop oldest work
                    pass
        # periodic logs (reference push reward_logs / success_rate)
        if log_queue is not None:
            for t in set(task_idx_list):
                done_eps = len(rollout.episode_rewards[t])
                if print_period_episodes and \
                        done_eps >= episodes_seen[t] + print_period_episodes:
                    episodes_seen[t] = done_eps
                    recent = rollout.episode_rewards[t][-print_period_episodes:]
                    log_queue.put(("reward", player_id, t,
                                   rollout.total_steps_per_task[t],
                                   float(np.mean(recent))))
                if eval_every_episodes and \
                        done_eps >= eval_done_at[t] + eval_every_episodes:
                    eval_done_at[t] = done_eps
                    rate = rollout.evaluate_success_rate(t, episodes=10)
                    log_queue.put(("success_rate", player_id, t,
                                   rollout.total_steps_per_task[t], rate))
        n_chunks += 1
        if max_chunks is not None and n_chunks >= max_chunks:
            break


@torch.no_grad()
def evaluate_checkpoint(cfg: SACConfig, checkpoint_path: str, env_fn,
                        task_idx: int = 0, episodes: int = 50,
                        render: bool = False, seed: int = 0) -> Dict:
    """Reference eval mode: deterministic policy from a .tar checkpoint,
    success-rate/reward over N episodes (main.py else-branch +
    player.calculate_success_rate)."""
    actor = build_actor(cfg)
    if cfg.variant == "care":
        from ..checkpoint import load_checkpoint
        ckpt = load_checkpoint(checkpoint_path)
        actor.load_state_dict_from_checkpoint(ckpt)
        it = int(ckpt.get("update_iteration", 0))
    else:
        it = load_actor_for_eval(actor, checkpoint_path)
    actor.eval()
    env = env_fn(cfg, task_idx, seed)
    rewards, successes = [], 0
    for ep in range(episodes):
        state = env.reset()
        total, succeeded = 0.0, False
        for _ in range(cfg.max_episode_time):
            obs = state
            if cfg.variant in ("mtsac", "care"):
                oh = np.zeros(cfg.num_tasks, dtype=np.float32)
                oh[task_idx] = 1.0
                obs = np.concatenate([state, oh])
            x = torch.from_numpy(np.asarray(obs, dtype=np.float32)[None])
            a = actor.get_action(x, stochastic=False).numpy()[0]
            state, r, done, info = env.step(a)
            total += r
            if render:
                env.render()
            if info.get("success", 0):
                succeeded = True
                break
            if done:
                break
        rewards.append(total)
        successes += int(succeeded)
    return {"update_iteration": it, "episodes": episodes,
            "mean_reward": float(np.mean(rewards)),
            "success_rate": successes / episodes}
