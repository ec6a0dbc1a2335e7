"""Single-process trainer: vectorized rollout + learner in one loop.

The minimum end-to-end slice: a synthetic (or real) env
set feeds the GPU-resident replay, the SACEngine updates, and weights are
"published" by construction (rollout shares the learner's actor).  The
fully asynchronous multi-process topology lives in
:mod:`distributed_sac_amd.workers.player` / ``.learner``.
"""

from __future__ import annotations

from typing import Callable, Dict, Optional

import torch

from ..algo import create_engine
from ..config import SACConfig
from ..replay import ShardedReplay
from ..utils import MetricLogger, StepTimer
from .rollout import VecRollout


def default_env_fn(cfg: SACConfig, task_idx: int, seed: int):
    from ..envs.synthetic import SyntheticEnv
    return SyntheticEnv(cfg.state_dim, cfg.action_dim,
                        max_episode_steps=cfg.max_episode_time, seed=seed,
                        success_info=cfg.variant in ("vsac", "mtsac", "care"),
                        action_bound=tuple(cfg.action_bound),
                        dynamics_seed=task_idx)   # same task = same system


class _CAREEnginePolicy:
    """In-process rollout adapter: engine's own context encoder + actor
    (single-process Trainer shares weights by construction)."""

    def __init__(self, engine):
        self.engine = engine

    @torch.no_grad()
    def get_action(self, mtobss, stochastic=True):
        z = self.engine.context_encoder(mtobss)
        return self.engine.actor.get_action(mtobss, z, stochastic=stochastic)


class Trainer:
    def __init__(self, cfg: SACConfig, device: str = "cpu",
                 env_fn: Optional[Callable] = None,
                 envs_per_task: int = 1, seed: int = 0,
                 logger: Optional[MetricLogger] = None):
        self.cfg = cfg
        self.device = torch.device(device)
        if self.device.type != "cuda":
            torch.set_num_threads(1)   # tiny ops thrash the intra-op pool
        self.engine = create_engine(cfg, device)
        num_tasks = cfg.num_tasks if cfg.variant in ("mtsac", "care") else 1
        self.replay = ShardedReplay(cfg.buffer_size, num_tasks,
                                    cfg.mtobs_dim, cfg.action_dim,
                                    device=device, seed=seed)
        env_fn = env_fn or default_env_fn
        envs, tasks = [], []
        for t in range(num_tasks):
            for j in range(envs_per_task):
                envs.append(env_fn(cfg, t, seed * 10007 + t * 101 + j))
                tasks.append(t)
        rollout_actor = self.engine.actor
        if cfg.variant == "care":
            rollout_actor = _CAREEnginePolicy(self.engine)
        self.rollout = VecRollout(cfg, envs, tasks, rollout_actor,
                                  device=device, seed=seed)
        self.logger = logger or MetricLogger(None)
        self.env_timer = StepTimer()
        self.update_timer = StepTimer()

    def collect_steps(self, n_steps: int) -> int:
        blocks = self.rollout.collect(n_steps)
        pushed = 0
        for t, blk in blocks.items():
            n = blk["states"].shape[0]
            if n:
                self.replay.append_numpy(task_idx=t, **blk)
                pushed += n
        self.engine.total_step += pushed
        self.env_timer.mark(pushed)
        return pushed

    def ready(self) -> bool:
        return len(self.replay) >= min(self.cfg.start_memory_len,
                                       self.replay.shards[0].capacity)

    def update_once(self) -> Dict[str, float]:
        batch = self.replay.sample(self.cfg.batch_size)
        metrics = self.engine.update(batch)
        self.update_timer.mark()
        return metrics

    def train(self, env_steps_per_iter: int, updates_per_iter: int,
              iterations: int, log_every: int = 0) -> Dict[str, float]:
        metrics: Dict[str, float] = {}
        for i in range(iterations):
            self.collect_steps(env_steps_per_iter)
            if self.ready():
                for _ in range(updates_per_iter):
                    metrics = self.update_once()
            if log_every and i % log_every == 0 and metrics:
                self.logger.add_scalars("learner", metrics,
                                        self.engine.update_iteration)
        return metrics
