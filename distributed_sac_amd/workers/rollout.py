"""Vectorized rollout core — the Player's environment loop.

Re-design of the reference Player (LunarLander_Distributed_SAC/src/
player.py:13-153; MT10_Distributed_MTSAC/src/player.py:13-286): one worker
owns a SET of envs (possibly spanning several tasks, reference task
round-robin player.py:247-253) and steps them in lockstep with BATCHED
actor inference — instead of the reference's one-env-per-process B=1
inference + per-step Redis weight download.

Semantics kept from the reference:
- per-task random warmup for ``random_step`` env steps (player.py:189);
- mtobs = concat(state, one_hot(task)) (MT10…MTSAC/src/player.py:155-170);
- done-masking: LunarLander stores done=False when the episode hit the time
  limit... (reference player.py:111-112 masks `done` given max steps);
- success-rate evaluation: 50 deterministic episodes per task
  (MT1_Distributed_VSAC/src/player.py:102-143).
"""

from __future__ import annotations

from typing import Dict, List, Optional

import numpy as np
import torch

from ..config import SACConfig


def one_hot(idx: int, n: int) -> np.ndarray:
    v = np.zeros(n, dtype=np.float32)
    v[idx] = 1.0
    return v


class VecRollout:
    """Steps N envs with batched inference on a local actor copy."""

    def __init__(self, cfg: SACConfig, envs: List, task_indices: List[int],
                 actor: torch.nn.Module, device: torch.device | str = "cpu",
                 seed: int = 0):
        assert len(envs) == len(task_indices)
        self.cfg = cfg
        self.envs = envs
        self.task_indices = task_indices
        self.actor = actor
        self.device = torch.device(device)
        self.num_tasks = cfg.num_tasks if cfg.variant in ("mtsac", "care") else 0
        self.rng = np.random.default_rng(seed)
        self.obs = [self._mtobs(e.reset(), t) for e, t in zip(envs, task_indices)]
        self.ep_steps = [0] * len(envs)
        self.ep_rewards = [0.0] * len(envs)
        self.total_steps_per_task: Dict[int, int] = {t: 0 for t in set(task_indices)}
        self.episode_rewards: Dict[int, List[float]] = {t: [] for t in set(task_indices)}
        self.warmup_remaining = {t: cfg.random_step for t in set(task_indices)}

    def _mtobs(self, state: np.ndarray, task_idx: int) -> np.ndarray:
        if self.num_tasks:
            return np.concatenate([state, one_hot(task_idx, self.num_tasks)])
        return np.asarray(state, dtype=np.float32)

    @torch.no_grad()
    def _policy_actions(self, obs_batch: np.ndarray) -> np.ndarray:
        x = torch.from_numpy(obs_batch).to(self.device)
        a = self.actor.get_action(x, stochastic=True)
        return a.cpu().numpy()

    def collect(self, n_steps: int) -> Dict[int, Dict[str, np.ndarray]]:
        """Run n_steps lockstep env steps; returns per-task transition blocks
        {task: {states, actions, rewards, next_states, dones}}."""
        out: Dict[int, Dict[str, List]] = {
            t: {k: [] for k in ("states", "actions", "rewards",
                                "next_states", "dones")}
            for t in set(self.task_indices)}
        n_envs = len(self.envs)
        for _ in range(n_steps):
            obs_batch = np.stack(self.obs).astype(np.float32)
            need_policy = [self.warmup_remaining[t] <= 0 for t in self.task_indices]
            actions = np.zeros((n_envs, self.cfg.action_dim), dtype=np.float32)
            if any(need_policy):
                pol = self._policy_actions(obs_batch)
                for i, np_ in enumerate(need_policy):
                    if np_:
                        actions[i] = pol[i]
            for i, np_ in enumerate(need_policy):
                if not np_:
                    actions[i] = self.envs[i].action_space.sample()

            for i, env in enumerate(self.envs):
                t = self.task_indices[i]
                next_state, reward, done, info = env.step(actions[i])
                self.ep_steps[i] += 1
                self.ep_rewards[i] += reward
                next_obs = self._mtobs(next_state, t)
                # time-limit masking (reference player.py stores done=False
                # on max_episode_time truncation so bootstrap continues)
                timeout = self.ep_steps[i] >= self.cfg.max_episode_time
                stored_done = bool(done) and not timeout
                out[t]["states"].append(self.obs[i])
                out[t]["actions"].append(actions[i])
                out[t]["rewards"].append(reward)
                out[t]["next_states"].append(next_obs)
                out[t]["dones"].append(float(stored_done))
                if self.warmup_remaining[t] > 0:
                    self.warmup_remaining[t] -= 1
                self.total_steps_per_task[t] += 1
                if done or timeout:
                    self.episode_rewards[t].append(self.ep_rewards[i])
                    self.obs[i] = self._mtobs(env.reset(), t)
                    self.ep_steps[i] = 0
                    self.ep_rewards[i] = 0.0
                else:
                    self.obs[i] = next_obs
        return {t: {k: np.asarray(v, dtype=np.float32)
                    for k, v in blk.items()} for t, blk in out.items()}

    @torch.no_grad()
    def evaluate_success_rate(self, task_idx: int, episodes: int = 50,
                              max_steps: Optional[int] = None) -> float:
        """Deterministic 50-episode protocol (reference
        MT1_Distributed_VSAC/src/player.py:102-143)."""
        env = self.envs[self.task_indices.index(task_idx)]
        max_steps = max_steps or self.cfg.max_episode_time
        successes = 0
        for _ in range(episodes):
            state = env.reset()
            obs = self._mtobs(state, task_idx)
            succeeded = False
            for _ in range(max_steps):
                x = torch.from_numpy(obs[None].astype(np.float32)).to(self.device)
                a = self.actor.get_action(x, stochastic=False).cpu().numpy()[0]
                state, reward, done, info = env.step(a)
                obs = self._mtobs(state, task_idx)
                if info.get("success", 0):
                    succeeded = True
                    break
                if done:
                    break
            successes += int(succeeded)
        # restore training episode state for that env
        i = self.task_indices.index(task_idx)
        self.obs[i] = self._mtobs(env.reset(), task_idx)
        self.ep_steps[i] = 0
        self.ep_rewards[i] = 0.0
        return successes / episodes


class FastSyntheticRollout:
    """High-throughput rollout over BatchedSyntheticEnv: one numpy step +
    one batched policy inference per lockstep tick across all tasks.

    Same contract as VecRollout.collect (per-task transition blocks with
    warmup and mtobs semantics); terminal flags are pure time-limits in the
    synthetic family, so stored dones are 0 (reference time-limit masking).
    """

    def __init__(self, cfg: SACConfig, task_indices: List[int],
                 actor: torch.nn.Module, envs_per_task: int = 32,
                 device: torch.device | str = "cpu", seed: int = 0):
        from ..envs.synthetic import BatchedSyntheticEnv
        self.cfg = cfg
        self.tasks = sorted(set(task_indices))
        self.actor = actor
        self.device = torch.device(device)
        self.num_tasks = cfg.num_tasks if cfg.variant in ("mtsac", "care") else 0
        self.envs_per_task = envs_per_task
        self.envs: Dict[int, object] = {}
        self.obs: Dict[int, np.ndarray] = {}
        for t in self.tasks:
            env = BatchedSyntheticEnv(
                envs_per_task, cfg.state_dim, cfg.action_dim,
                max_episode_steps=cfg.max_episode_time,
                seed=seed * 9173 + t,
                success_info=cfg.variant in ("vsac", "mtsac", "care"),
                action_bound=tuple(cfg.action_bound),
                dynamics_seed=t)   # same task = same system
            self.envs[t] = env
            self.obs[t] = env.reset_all()
        self.warmup_remaining = {t: cfg.random_step for t in self.tasks}
        self.total_steps_per_task = {t: 0 for t in self.tasks}
        # one-hot suffix per task, precomputed
        self._oh = {t: one_hot(t, self.num_tasks) if self.num_tasks else None
                    for t in self.tasks}

    def _mtobs_block(self, t: int, states: np.ndarray) -> np.ndarray:
        if not self.num_tasks:
            return states
        oh = np.broadcast_to(self._oh[t], (states.shape[0], self.num_tasks))
        return np.concatenate([states, oh], axis=1)

    @torch.no_grad()
    def collect(self, n_steps: int) -> Dict[int, Dict[str, np.ndarray]]:
        nt, ne = len(self.tasks), self.envs_per_task
        S = self.cfg.mtobs_dim
        A = self.cfg.action_dim
        out = {t: {k: [] for k in FIELDS_ROLLOUT} for t in self.tasks}
        for _ in range(n_steps):
            obs_all = np.concatenate(
                [self._mtobs_block(t, self.obs[t]) for t in self.tasks])
            # batched policy inference across every env of every task
            any_policy = any(self.warmup_remaining[t] <= 0 for t in self.tasks)
            if any_policy:
                x = torch.from_numpy(obs_all.astype(np.float32)).to(self.device)
                pol = self.actor.get_action(x, stochastic=True).cpu().numpy()
            for i, t in enumerate(self.tasks):
                env = self.envs[t]
                if self.warmup_remaining[t] > 0:
                    acts = env.sample_actions()
                    self.warmup_remaining[t] -= ne
                else:
                    acts = pol[i * ne:(i + 1) * ne]
                states_mt = obs_all[i * ne:(i + 1) * ne]
                next_states, rewards, dones, _succ = env.step_all(acts)
                next_mt = self._mtobs_block(t, next_states)
                out[t]["states"].append(states_mt)
                out[t]["actions"].append(acts)
                out[t]["rewards"].append(rewards.reshape(-1, 1))
                out[t]["next_states"].append(next_mt)
                # synthetic terminals are pure time-limits -> masked to 0
                out[t]["dones"].append(np.zeros((ne, 1), dtype=np.float32))
                self.obs[t] = env.states.copy()
                self.total_steps_per_task[t] += ne
        return {t: {k: np.concatenate(v).astype(np.float32)
                    for k, v in blk.items()} for t, blk in out.items()}


FIELDS_ROLLOUT = ("states", "actions", "rewards", "next_states", "dones")
