"""Env registry: synthetic always; gym / Meta-World adapters when installed.

The reference builds envs in each variant's main.py (``gym.make('LunarLander
Continuous-v2')``; ``metaworld.MT1/MT10`` train_classes/tasks).  Those
libraries are optional here: :func:`make_env` returns the real env when the
import succeeds and the synthetic stand-in otherwise (CI has neither).
"""

from __future__ import annotations

from .synthetic import BoxSpace, SyntheticEnv, make_synthetic  # noqa: F401


def have_gym() -> bool:
    try:
        import gym  # noqa: F401
        return True
    except ImportError:
        return False


def have_metaworld() -> bool:
    try:
        import metaworld  # noqa: F401
        return True
    except ImportError:
        return False


def make_env(name: str, seed: int = 0, allow_synthetic: bool = True):
    """name: 'LunarLanderContinuous-v2', 'mt1-<task>', 'mt10', or
    'synthetic-lunarlander' / 'synthetic-metaworld'."""
    if name.startswith("synthetic-"):
        return make_synthetic(name.split("-", 1)[1], seed)
    if name.startswith("LunarLander"):
        if have_gym():
            import gym
            env = gym.make(name)
            env.seed(seed)
            return env
        if allow_synthetic:
            return make_synthetic("lunarlander", seed)
        raise ImportError("gym not installed and allow_synthetic=False")
    if name.startswith("mt1") or name == "mt10":
        if have_metaworld():
            return _make_metaworld(name, seed)
        if allow_synthetic:
            return make_synthetic("metaworld", seed)
        raise ImportError("metaworld not installed and allow_synthetic=False")
    raise ValueError(f"unknown env {name}")


class _MT1EpisodeTaskEnv:  # pragma: no cover - needs metaworld
    """Reference MT1 players call ``set_task(random.choice(train_tasks))``
    before every episode (MT1_Distributed_VSAC/src/player.py:93-100) —
    re-randomize the goal on each reset so rollouts see the full task
    distribution, not one frozen goal."""

    def __init__(self, env, train_tasks, seed: int):
        import random
        self._env = env
        self._tasks = list(train_tasks)
        self._rng = random.Random(seed)

    def reset(self):
        self._env.set_task(self._rng.choice(self._tasks))
        return self._env.reset()

    def __getattr__(self, name):
        return getattr(self._env, name)


def _make_metaworld(name: str, seed: int):  # pragma: no cover - needs metaworld
    """Meta-World construction mirroring reference main.py (MT1: env from
    train_classes + per-episode set_task over train_tasks; MT10: dict of
    classes)."""
    import metaworld
    if name == "mt10":
        return metaworld.MT10()
    task_name = name.split("-", 1)[1] if "-" in name else "pick-place-v2"
    mt1 = metaworld.MT1(task_name)
    env = mt1.train_classes[task_name]()
    return _MT1EpisodeTaskEnv(env, mt1.train_tasks, seed)
