"""Synthetic environments — the CI/bench workhorse.

There is no gym/metaworld (and no network) in the build image, and
BASELINE.json specifies benches on "synthetic transitions / random-init
weights".  These envs expose the pre-gym-0.26 API the reference codes
against (``step() -> (obs, reward, done, info)``; reference player.py) with
the exact observation/action shapes of the reference tasks:

- :class:`SyntheticEnv` — seeded random linear dynamics, bounded state,
  quadratic reward; arbitrary dims.
- :func:`make_synthetic` — shape presets: ``lunarlander`` (8/2),
  ``metaworld`` (39/4, with a Meta-World-style ``info['success']`` flag).
"""

from __future__ import annotations

from typing import Optional, Tuple

import numpy as np


class BoxSpace:
    """Minimal gym.spaces.Box stand-in (sample/shape only)."""

    def __init__(self, low: float, high: float, shape, rng: np.random.Generator):
        self.low, self.high, self.shape = low, high, tuple(shape)
        self._rng = rng

    def sample(self) -> np.ndarray:
        return self._rng.uniform(self.low, self.high, self.shape).astype(np.float32)


class SyntheticEnv:
    """Seeded random linear-dynamics env with the pre-0.26 gym API."""

    def __init__(self, state_dim: int, action_dim: int,
                 max_episode_steps: int = 500, seed: int = 0,
                 success_info: bool = False, action_bound=(-1.0, 1.0),
                 dynamics_seed: Optional[int] = None):
        self.state_dim = state_dim
        self.action_dim = action_dim
        self.max_episode_steps = max_episode_steps
        self.success_info = success_info
        self._rng = np.random.default_rng(seed)
        # fixed, well-conditioned dynamics: s' = 0.98*A s + B a + noise.
        # dynamics_seed decouples the SYSTEM from the episode randomness:
        # every worker playing "task t" must face the SAME dynamics (a
        # per-worker seed only varies resets/noise) — otherwise one task's
        # replay shard mixes transitions from different systems.
        rng_dyn = np.random.default_rng(
            seed if dynamics_seed is None else dynamics_seed)
        self.A = rng_dyn.normal(0, 1.0 / np.sqrt(state_dim),
                                (state_dim, state_dim)).astype(np.float32)
        self.B = rng_dyn.normal(0, 0.3, (state_dim, action_dim)).astype(np.float32)
        self.action_space = BoxSpace(action_bound[0], action_bound[1],
                                     (action_dim,), self._rng)
        self.observation_space = BoxSpace(-np.inf, np.inf, (state_dim,), self._rng)
        self._t = 0
        self._state = np.zeros(state_dim, dtype=np.float32)

    def seed(self, seed: int) -> None:
        self._rng = np.random.default_rng(seed)
        self.action_space._rng = self._rng

    def reset(self) -> np.ndarray:
        self._t = 0
        self._state = self._rng.normal(0, 1, self.state_dim).astype(np.float32)
        return self._state.copy()

    def step(self, action) -> Tuple[np.ndarray, float, bool, dict]:
        a = np.clip(np.asarray(action, dtype=np.float32).reshape(-1),
                    self.action_space.low, self.action_space.high)
        noise = self._rng.normal(0, 0.05, self.state_dim).astype(np.float32)
        self._state = 0.98 * (self.A @ self._state) + self.B @ a + noise
        self._state = np.clip(self._state, -10.0, 10.0)
        self._t += 1
        reward = float(-0.1 * np.square(self._state).mean()
                       - 0.01 * np.square(a).mean())
        done = self._t >= self.max_episode_steps
        info = {}
        if self.success_info:
            # Meta-World-style success flag (reference player.is_success
            # reads info['success'])
            info["success"] = float(np.square(self._state).mean() < 0.5)
        return self._state.copy(), reward, done, info

    def render(self, *a, **k):  # pragma: no cover - no-op
        pass

    def close(self):  # pragma: no cover
        pass

    # Meta-World MT1-style task switching API (env.set_task)
    def set_task(self, task) -> None:
        seed = task if isinstance(task, (int, np.integer)) else hash(task) % (2**31)
        self.seed(int(seed))


def make_synthetic(preset: str = "lunarlander", seed: int = 0,
                   max_episode_steps: Optional[int] = None,
                   dynamics_seed: Optional[int] = None) -> SyntheticEnv:
    if preset == "lunarlander":
        return SyntheticEnv(8, 2, max_episode_steps or 500, seed,
                            dynamics_seed=dynamics_seed)
    if preset == "metaworld":
        return SyntheticEnv(39, 4, max_episode_steps or 500, seed,
                            success_info=True, dynamics_seed=dynamics_seed)
    raise ValueError(preset)


class BatchedSyntheticEnv:
    """N synthetic envs stepped as ONE numpy batch op.

    The rollout worker's lockstep loop (VecRollout) spends most of its CPU
    time in per-env python ``step`` calls; this env family steps all N
    copies in one vectorized update (states [N, S]), lifting a worker's
    env-steps/s by an order of magnitude.  Same dynamics family as
    :class:`SyntheticEnv` (shared A/B per instance, per-env noise).
    """

    def __init__(self, n_envs: int, state_dim: int, action_dim: int,
                 max_episode_steps: int = 500, seed: int = 0,
                 success_info: bool = False, action_bound=(-1.0, 1.0),
                 dynamics_seed: Optional[int] = None):
        self.n_envs = n_envs
        self.state_dim = state_dim
        self.action_dim = action_dim
        self.max_episode_steps = max_episode_steps
        self.success_info = success_info
        self._rng = np.random.default_rng(seed)
        rng_dyn = np.random.default_rng(
            seed if dynamics_seed is None else dynamics_seed)
        self.A = rng_dyn.normal(0, 1.0 / np.sqrt(state_dim),
                                (state_dim, state_dim)).astype(np.float32)
        self.B = rng_dyn.normal(0, 0.3, (state_dim, action_dim)).astype(np.float32)
        self.low, self.high = action_bound
        self.states = np.zeros((n_envs, state_dim), dtype=np.float32)
        self.t = np.zeros(n_envs, dtype=np.int64)

    def reset_all(self) -> np.ndarray:
        self.t[:] = 0
        self.states = self._rng.normal(
            0, 1, (self.n_envs, self.state_dim)).astype(np.float32)
        return self.states.copy()

    def sample_actions(self) -> np.ndarray:
        return self._rng.uniform(self.low, self.high,
                                 (self.n_envs, self.action_dim)).astype(np.float32)

    def step_all(self, actions: np.ndarray):
        """Returns (next_states, rewards, dones, successes); auto-resets
        finished envs (dones reflect the pre-reset transition)."""
        a = np.clip(actions.astype(np.float32), self.low, self.high)
        noise = self._rng.normal(0, 0.05,
                                 self.states.shape).astype(np.float32)
        self.states = 0.98 * (self.states @ self.A.T) + a @ self.B.T + noise
        np.clip(self.states, -10.0, 10.0, out=self.states)
        self.t += 1
        rewards = (-0.1 * np.square(self.states).mean(axis=1)
                   - 0.01 * np.square(a).mean(axis=1)).astype(np.float32)
        dones = self.t >= self.max_episode_steps
        success = (np.square(self.states).mean(axis=1) < 0.5).astype(np.float32) \
            if self.success_info else np.zeros(self.n_envs, dtype=np.float32)
        out_states = self.states.copy()
        if dones.any():
            idx = np.nonzero(dones)[0]
            self.states[idx] = self._rng.normal(
                0, 1, (len(idx), self.state_dim)).astype(np.float32)
            self.t[idx] = 0
        return out_states, rewards, dones, success
