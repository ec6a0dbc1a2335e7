"""Orchestrator — spawns player processes around a learner.

Replaces the reference main.py + Ray topology (LunarLander_Distributed_
SAC/src/main.py:16-45 etc.): N player processes (torch.multiprocessing
spawn) push transition blocks into a queue; the learner (this process)
ingests, updates on the GPU, and publishes weights through the
shared-memory snapshot.  No Redis, no Ray, no fractional-GPU time-sharing
— players are CPU-only; the learner owns the GPU.
"""

from __future__ import annotations

import time
from typing import Callable, Dict, List, Optional

import torch
import torch.multiprocessing as mp

from ..config import SACConfig
from ..utils import MetricLogger
from .learner import Learner
from .param_server import ParamSnapshot
from .player import run_player
from .trainer import default_env_fn


def _actor_numel(cfg: SACConfig) -> int:
    from .player import build_actor, policy_params
    return sum(p.numel() for p in policy_params(build_actor(cfg)))


def default_task_partition(num_tasks: int, num_players: int) -> List[List[int]]:
    """Round-robin tasks over players (reference MT10 main.py task
    partition lists, main.py:26-31)."""
    parts: List[List[int]] = [[] for _ in range(num_players)]
    for t in range(num_tasks):
        parts[t % num_players].append(t)
    for i, p in enumerate(parts):
        if not p:
            p.append(i % max(1, num_tasks))
    return parts


class DistributedTrainer:
    """num_players rollout processes + in-process learner."""

    def __init__(self, cfg: SACConfig, device: str = "cpu",
                 num_players: int = 2, env_fn: Callable = None,
                 logger: Optional[MetricLogger] = None,
                 save_dir: Optional[str] = None, save_period: int = 0,
                 chunk_steps: int = 64, seed: int = 0, use_graph: bool = True,
                 ddp=None, transport: str = "auto", precision: str = None,
                 eval_every_episodes: int = 0):
        self.cfg = cfg
        self.env_fn = env_fn or default_env_fn
        self.ctx = mp.get_context("spawn")
        self.snapshot = ParamSnapshot(_actor_numel(cfg))
        self.sample_queue = self.ctx.Queue(maxsize=1024)
        self.log_queue = self.ctx.Queue(maxsize=4096)
        self.stop_event = self.ctx.Event()
        self.heartbeat = torch.zeros(num_players + 1, dtype=torch.float64)
        self.heartbeat.share_memory_()
        num_tasks = cfg.num_tasks if cfg.variant in ("mtsac", "care") else 1
        self.partitions = default_task_partition(num_tasks, num_players)
        self.players: List[mp.Process] = []
        self._respawns = {}
        self.chunk_steps = chunk_steps
        self.seed = seed
        # reference Player(eval_episode_idx=40): periodic deterministic
        # success-rate evaluation in the players (0 = off)
        self.eval_every_episodes = eval_every_episodes
        # native shared-memory SPSC rings (one per player) when available;
        # mp.Queue otherwise ("auto"), or forced via transport=
        from .. import ops as _ops
        use_rings = (transport == "shm"
                     or (transport == "auto" and _ops.has_native()))
        self.rings = []
        self.ring_names: List[str] = []
        if use_rings:
            import os as _os
            ext = _ops.native()
            slot_floats = chunk_steps * (2 * cfg.mtobs_dim
                                         + cfg.action_dim + 2) + 64
            for pid in range(num_players):
                name = f"/dsac_{_os.getpid()}_{seed}_{pid}"
                self.rings.append(ext.ShmRing(name, 256, slot_floats,
                                              cfg.mtobs_dim, cfg.action_dim))
                self.ring_names.append(name)
        self.learner = Learner(cfg, device, self.snapshot, self.sample_queue,
                               self.log_queue, logger=logger,
                               save_dir=save_dir, save_period=save_period,
                               use_graph=use_graph, ddp=ddp, seed=seed,
                               heartbeat=self.heartbeat, rings=self.rings,
                               precision=precision)

    def _spawn_player(self, pid: int, seed_bump: int = 0) -> "mp.Process":
        p = self.ctx.Process(
            target=run_player,
            args=(pid, self.cfg, self.env_fn, self.partitions[pid],
                  self.snapshot, self.sample_queue, self.log_queue,
                  self.stop_event, self.chunk_steps,
                  self.seed + 131 * pid + 7907 * seed_bump, 2,
                  self.eval_every_episodes, None, self.heartbeat,
                  self.ring_names[pid] if self.ring_names else None),
            daemon=True)
        p.start()
        return p

    def start_players(self) -> None:
        for pid in range(len(self.partitions)):
            self.players.append(self._spawn_player(pid))
        # recovery (VERDICT round-1 item 8): respawn heartbeat-dead
        # players on the SAME ring, bounded per player
        self.learner.on_dead_player = self.respawn_player

    def respawn_player(self, pid: int, max_respawns: int = 5) -> bool:
        """Replace a dead player process, reusing its ring/partition.
        Returns True when a fresh process is running."""
        if pid >= len(self.players):
            return False
        n = self._respawns.get(pid, 0)
        if n >= max_respawns:
            return False
        old = self.players[pid]
        if old.is_alive():
            old.terminate()
        old.join(timeout=5.0)
        self._respawns[pid] = n + 1
        self.players[pid] = self._spawn_player(pid, seed_bump=n + 1)
        return True

    def run(self, max_grad_steps: Optional[int] = None,
            max_seconds: Optional[float] = None) -> Dict[str, float]:
        self.start_players()
        try:
            stats = self.learner.run(stop_event=self.stop_event,
                                     max_grad_steps=max_grad_steps,
                                     max_seconds=max_seconds)
        finally:
            self.shutdown()
        return stats

    def shutdown(self, timeout: float = 10.0) -> None:
        self.stop_event.set()
        # drain so players blocked on a full queue can exit
        deadline = time.time() + timeout
        for p in self.players:
            while p.is_alive() and time.time() < deadline:
                try:
                    while True:
                        self.sample_queue.get_nowait()
                except Exception:
                    pass
                p.join(timeout=0.2)
            if p.is_alive():
                p.terminate()
                p.join(timeout=2.0)
        self.players = []
