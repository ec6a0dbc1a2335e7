"""Learner process — replay ingest + SAC updates + weight publishing.

Re-implements the reference Learner process (LunarLander_Distributed_SAC/
src/learner.py:21-316, MT10_Distributed_MTSAC/src/learner.py) without
Redis: transitions arrive as blocks on a multiprocessing queue and go
straight into the HBM-resident sharded replay; weights publish through the
shared-memory seqlock snapshot (one D2H copy per publish, no pickling).

Reference semantics kept:
- wait until ``len(replay) >= start_memory_len`` before training
  (learner.wait_until_memoryReady);
- hard copy critics -> targets at train start (soft_update tau=1.0);
- ``update_delay`` iteration thinning: the published update_iteration
  counter advances every loop but a gradient update runs only when
  ``it % update_delay == 0`` (learner.run:293-295) — so published
  iteration = grad_steps * update_delay;
- checkpoint every ``save_period`` real updates.
"""

from __future__ import annotations

import os

import queue as pyqueue
import time
from typing import Dict, Optional

import torch

from ..algo import create_engine
from ..checkpoint import save_checkpoint
from ..config import SACConfig
from ..replay import ShardedReplay
from ..utils import MetricLogger, StepTimer
from .param_server import ParamSnapshot


class Learner:
    def __init__(self, cfg: SACConfig, device: str, snapshot: ParamSnapshot,
                 sample_queue, log_queue=None,
                 logger: Optional[MetricLogger] = None,
                 save_dir: Optional[str] = None, save_period: int = 0,
                 update_delay: Optional[int] = None, use_graph: bool = True,
                 ddp=None, seed: int = 0, heartbeat=None,
                 heartbeat_timeout: float = 60.0, rings=None,
                 precision: str = None, publish_interval_s: float = 0.02,
                 graph_chunk: int = 4):
        self.cfg = cfg
        self.device = torch.device(device)
        if self.device.type != "cuda":
            # tiny per-op work: the default intra-op pool THRASHES (measured
            # 478 ms vs 4.7 ms per update for a 64-wide net on 8 cores)
            torch.set_num_threads(1)
        if precision is None:
            precision = "bf16" if torch.device(device).type == "cuda" \
                else "fp32"
        self.engine = create_engine(cfg, device, precision=precision)
        if ddp is not None:
            self.engine.attach_ddp(ddp)
        num_tasks = cfg.num_tasks if cfg.variant in ("mtsac", "care") else 1
        self.replay = ShardedReplay(cfg.buffer_size, num_tasks,
                                    cfg.mtobs_dim, cfg.action_dim,
                                    device=device, seed=seed)
        self.snapshot = snapshot
        self.sample_queue = sample_queue
        self.rings = rings or []
        if self.rings and self.device.type == "cuda":
            for r in self.rings:
                r.pin()  # hipHostRegister: DMA-able drains
        self.log_queue = log_queue
        self.logger = logger or MetricLogger(None)
        self.save_dir = save_dir
        self.save_period = save_period
        self.update_delay = (update_delay if update_delay is not None
                             else max(1, cfg.update_delay))
        dp_active = ddp is not None and ddp.enabled
        self.use_graph = (use_graph and self.device.type == "cuda"
                          and not dp_active)
        # DP learners use the segmented capture (three hipGraphs, eager
        # RCCL all-reduces between replays — SACEngine.capture_dp)
        self.use_dp_graph = (use_graph and self.device.type == "cuda"
                             and dp_active
                             and getattr(self.engine, "_bf16", False))
        self._graph_ready = False
        self._dp_graph_ready = False
        self.iteration_counter = 0   # reference update_iteration (thinned)
        self.grad_steps = 0
        self.update_timer = StepTimer()
        self.ingest_count = 0
        # per-phase wall-time accounting (SURVEY §5.1 — absent in reference)
        # update/publish are sub-spans of loop (timed inside train_step);
        # other = heartbeat checks; loop - update - publish = python glue
        self.phase_seconds = {"drain": 0.0, "logs": 0.0, "update": 0.0,
                              "sync": 0.0, "publish": 0.0, "other": 0.0,
                              "loop": 0.0}
        self._pub_pinned = None
        # publish throttle: the learner updates orders of magnitude faster
        # than players poll; the reference's version-gated pull (apply only
        # when update_iteration changed) makes intermediate publishes
        # unobservable, so we publish at most every publish_interval_s
        # (docs/DESIGN_NOTES.md) instead of syncing D2H every update.
        self.publish_interval_s = float(
            os.environ.get("DSAC_PUBLISH_S", publish_interval_s))
        # updates captured per hipGraph (amortizes the ~0.22 ms host
        # launch; every update in the chunk is complete and distinct)
        self.graph_chunk = max(1, int(graph_chunk))
        self._metric_every = int(os.environ.get("DSAC_METRICS_EVERY", 100))
        self._met_pinned = None
        self._met_event = None
        self._met_inflight = False
        self._next_metric_tick = self._metric_every
        self._last_publish = 0.0
        self._pub_stream = None
        self._pub_event = None
        self._pub_iteration = 0
        # failure detection (absent in the reference — SURVEY §5.3):
        # shared wall-clock heartbeats, slot -1 = learner, others = players
        self.heartbeat = heartbeat
        self.heartbeat_timeout = heartbeat_timeout
        self._dead_players = set()
        self._hb_check = 0
        # recovery hook (round 2): the orchestrator owns
        # the process objects and ring names, so it installs a callback
        # that respawns a heartbeat-dead player on the SAME ring;
        # returning True clears the player's dead mark
        self.on_dead_player = None

    def check_heartbeats(self) -> None:
        if self.heartbeat is None:
            return
        self._hb_check += 1
        if self._hb_check % 200 != 0:
            return
        import time as _t
        now = _t.time()
        self.heartbeat[-1] = now
        for pid in range(self.heartbeat.numel() - 1):
            last = float(self.heartbeat[pid])
            if last > 0 and now - last > self.heartbeat_timeout \
                    and pid not in self._dead_players:
                self._dead_players.add(pid)
                self.logger.print(
                    f"WARNING: player {pid} heartbeat stale "
                    f"({now - last:.0f}s)")
                if self.on_dead_player is not None:
                    try:
                        if self.on_dead_player(pid):
                            self.heartbeat[pid] = now
                            self._dead_players.discard(pid)
                            self.logger.print(f"player {pid} respawned")
                        else:
                            self.logger.print(
                                f"player {pid} NOT respawned (limit) — "
                                "continuing without it")
                    except Exception as e:  # pragma: no cover
                        self.logger.print(f"respawn of {pid} failed: {e!r}")

    @property
    def dead_players(self):
        return set(self._dead_players)

    # -- ingest --------------------------------------------------------
    def drain_rings(self, max_blocks: int = 64) -> int:
        """Drain the native shared-memory SPSC rings (one per player)."""
        n = 0
        pinned = self.device.type == "cuda"
        for ring in self.rings:
            for _ in range(max_blocks):
                out = ring.pop(pinned)
                if not out:
                    break
                task = int(out[0].item())
                # pinned host tensors go straight into the shard's slice
                # copies (append does the async H2D DMA itself)
                self.replay.shards[task].append(out[1], out[2], out[3],
                                                out[4], out[5])
                got = out[1].shape[0]
                self.engine.total_step += got
                self.ingest_count += got
                n += 1
        return n

    def drain_queue(self, max_blocks: int = 64) -> int:
        n = self.drain_rings(max_blocks)
        for _ in range(max_blocks):
            try:
                _pid, task, blk = self.sample_queue.get_nowait()
            except (pyqueue.Empty, OSError):
                break
            self.replay.append_numpy(task_idx=task, **blk)
            got = blk["states"].shape[0]
            self.engine.total_step += got
            self.ingest_count += got
            n += 1
        return n

    def drain_logs(self) -> None:
        if self.log_queue is None:
            return
        for _ in range(256):
            try:
                item = self.log_queue.get_nowait()
            except (pyqueue.Empty, OSError):
                return
            kind, pid, task, step, value = item
            self.logger.add_scalar(f"{kind}/task_{task}_player_{pid}",
                                   value, step)

    # -- update --------------------------------------------------------
    def publish(self) -> None:
        """Two-phase asynchronous publish: the D2H copy runs on a side
        stream (never blocking the update pipeline); the shared-memory
        snapshot is written from the pinned staging buffer once the copy's
        event has completed (checked on the next publish tick).  Players
        see weights at most one publish interval stale — well inside the
        reference's own staleness (its players applied weights whenever
        the pickle download finished)."""
        flat = self.engine.publish_params()
        if not flat.is_cuda:
            self.snapshot.publish(flat, self.iteration_counter)
            return
        if self._pub_stream is None:
            self._pub_stream = torch.cuda.Stream(self.device)
            self._pub_event = torch.cuda.Event()
            self._pub_stage_event = torch.cuda.Event()
            self._pub_stage = torch.empty_like(flat.reshape(-1))
            self._pub_pinned = torch.empty(flat.numel(), pin_memory=True)
        # phase 1: flush the previous copy if it finished
        if self._pub_iteration and self._pub_event.query():
            self.snapshot.publish(self._pub_pinned, self._pub_iteration)
            self._pub_iteration = 0
        # phase 2: start a fresh async copy (skip if one is in flight).
        # Tear-safety: the D2D snapshot of flat_data is enqueued ON THE
        # MAIN stream, so it is ordered before any later-enqueued Adam
        # step that would overwrite flat_data — no main-stream wait_event
        # needed (the earlier stage-on-side-stream + main.wait design put
        # a full pipeline barrier on main at every publish tick, which
        # cost ~35% of the async rate).  The slow D2H then reads only the
        # immutable staging buffer on the side stream.
        if self._pub_iteration == 0:
            main = torch.cuda.current_stream(self.device)
            self._pub_stage.copy_(flat.reshape(-1), non_blocking=True)
            self._pub_stage_event.record(main)
            self._pub_stream.wait_event(self._pub_stage_event)
            with torch.cuda.stream(self._pub_stream):
                self._pub_pinned.copy_(self._pub_stage, non_blocking=True)
            self._pub_event.record(self._pub_stream)
            self._pub_iteration = self.iteration_counter

    def _metrics_tick(self, metrics_t) -> Optional[Dict[str, float]]:
        """Pipelined metric readback: D2H into pinned staging now, log
        the values once the copy's event has completed (next tick) —
        metrics are one tick stale but the learner never syncs."""
        if self._met_pinned is None:
            self._met_pinned = {k: torch.empty_like(v.detach(),
                                                    device="cpu",
                                                    pin_memory=True)
                                for k, v in metrics_t.items()}
            la = self.engine.log_alpha.detach()
            self._met_pinned["_log_alpha"] = torch.empty_like(
                la, device="cpu", pin_memory=True)
            self._met_event = torch.cuda.Event()
            self._met_inflight = False
        out = None
        if self._met_inflight and self._met_event.query():
            out = {k: float(v) for k, v in self._met_pinned.items()
                   if not k.startswith("_")}
            la = self._met_pinned["_log_alpha"]
            if la.numel() > 1:
                self.logger.add_scalars(
                    "alpha",
                    {f"task_{i}": float(v)
                     for i, v in enumerate(la.exp().tolist())},
                    self.grad_steps)
        for k, v in metrics_t.items():
            self._met_pinned[k].copy_(v.detach(), non_blocking=True)
        self._met_pinned["_log_alpha"].copy_(
            self.engine.log_alpha.detach(), non_blocking=True)
        self._met_event.record()
        self._met_inflight = True
        return out

    def ready(self) -> bool:
        """Reference start gate: train only once the MIN per-task shard
        holds ``start_memory_len`` transitions (MT10_Distributed_MTSAC/src/
        learner.py:354-358 with replay_buffers.__len__ = min over shards).
        ``len(self.replay)`` is already the min-shard count; capping by the
        shard capacity avoids a deadlock when a test-sized buffer is
        smaller than the gate.  Matches Trainer.ready."""
        return len(self.replay) > 0 and \
            len(self.replay) >= min(self.cfg.start_memory_len,
                                    self.replay.shards[0].capacity)

    def _ensure_graph(self) -> None:
        if self.use_graph and not self._graph_ready:
            try:
                self.logger.print("capturing update graph...")
                t0 = time.perf_counter()
                self.engine.capture(self.replay, self.cfg.batch_size,
                                    chunk=self.graph_chunk)
                self._graph_ready = True
                self.logger.print(
                    f"graph captured in {time.perf_counter() - t0:.1f}s")
            except Exception as e:  # pragma: no cover
                self.logger.print(f"hipGraph capture failed ({e!r}); eager")
                self.use_graph = False
        if self.use_dp_graph and not self._dp_graph_ready:
            try:  # pragma: no cover - requires multi-GPU RCCL
                self.logger.print("capturing segmented DP graphs...")
                t0 = time.perf_counter()
                self.engine.capture_dp(self.replay, self.cfg.batch_size)
                self._dp_graph_ready = True
                self.logger.print(
                    f"DP graphs captured in {time.perf_counter() - t0:.1f}s")
            except Exception as e:  # pragma: no cover
                self.logger.print(f"DP capture failed ({e!r}); eager")
                self.use_dp_graph = False

    def train_step(self) -> Optional[Dict[str, float]]:
        """One gradient update + its ``update_delay`` iteration-counter
        advance.  The reference loops ``update_delay`` times per update,
        doing nothing but counter++ on the thinned iterations
        (learner.run:293-295); at 1700+ updates/s those empty python
        iterations (plus their ring/log drains) cost ~30% of the async
        rate, so the counter advances in one step — the OBSERVABLE
        behavior (published update_iteration = grad_steps * update_delay,
        one drain per update) is unchanged."""
        metrics = None
        if True:
            t0 = time.perf_counter()
            if self.use_graph or self.use_dp_graph:
                self._ensure_graph()
            done = 1
            if self._graph_ready:
                out = self.engine.graphed_update()
                metrics_t = out
                done = getattr(self.engine, "_graph_chunk", 1)
            elif self._dp_graph_ready:  # pragma: no cover - multi-GPU
                metrics_t = self.engine.dp_graphed_update()
            else:
                metrics_t = self.engine.update_tensors(
                    self.replay.sample(self.cfg.batch_size,
                                       graph_safe=self.device.type == "cuda"))
                self.engine.update_iteration += 1
            self.iteration_counter += self.update_delay * done
            self.grad_steps += done
            self.update_timer.mark(done)
            t1 = time.perf_counter()
            self.phase_seconds["update"] += t1 - t0
            if t1 - self._last_publish >= self.publish_interval_s:
                self._last_publish = t1
                self.publish()   # pinned D2H copy syncs the queued updates
                self.phase_seconds["publish"] += time.perf_counter() - t1
            if self.save_dir and self.save_period and \
                    self.grad_steps % self.save_period < done:
                save_checkpoint(self.engine, self.save_dir,
                                update_iteration=self.iteration_counter)
            if self.grad_steps >= self._next_metric_tick:
                self._next_metric_tick = self.grad_steps + self._metric_every
                if self.device.type == "cuda":
                    # non-blocking tick: enqueue a D2H of the stat buffers
                    # into pinned staging and log the PREVIOUS tick's
                    # completed snapshot — float() here would sync the
                    # whole queued pipeline (~24% of the async rate)
                    metrics = self._metrics_tick(metrics_t)
                else:
                    metrics = {k: float(v) for k, v in metrics_t.items()}
                if metrics:
                    self.logger.add_scalars("learner", metrics,
                                            self.grad_steps)
                # per-task temperature curve (reference Logger writes the
                # full alpha array per update — MT10_Distributed_CARE/src/
                # logger.py:45-132, learner.py:445-464); one D2H of a
                # <=num_tasks vector per report tick
                if metrics and self.device.type != "cuda":
                    la = self.engine.log_alpha.detach()
                    if la.numel() > 1:
                        alphas = la.exp().cpu().tolist()
                        self.logger.add_scalars(
                            "alpha",
                            {f"task_{i}": float(v)
                             for i, v in enumerate(alphas)},
                            self.grad_steps)
        return metrics

    # -- main loop -----------------------------------------------------
    def run(self, stop_event=None, max_grad_steps: Optional[int] = None,
            max_seconds: Optional[float] = None) -> Dict[str, float]:
        self.publish()  # initial weights (reference learner.run start)
        t0 = time.perf_counter()
        while not self.ready():
            if stop_event is not None and stop_event.is_set():
                return {}
            self.drain_queue()
            self.drain_logs()
            time.sleep(0.01)
            if max_seconds and time.perf_counter() - t0 > max_seconds:
                return {}
        self.engine.hard_copy_targets()
        self.logger.print("######### Start train #########")
        last_report = time.perf_counter()
        while True:
            if stop_event is not None and stop_event.is_set():
                break
            p0 = time.perf_counter()
            self.drain_queue()
            p1 = time.perf_counter()
            self.phase_seconds["drain"] += p1 - p0
            self.drain_logs()
            p2 = time.perf_counter()
            self.phase_seconds["logs"] += p2 - p1
            self.check_heartbeats()
            p3 = time.perf_counter()
            self.phase_seconds["other"] += p3 - p2
            self.train_step()
            self.phase_seconds["loop"] += time.perf_counter() - p3
            if time.perf_counter() - last_report > 10.0:
                last_report = time.perf_counter()
                self.logger.print(
                    f"it={self.iteration_counter} grad={self.grad_steps} "
                    f"rate={self.update_timer.rate():.1f}/s "
                    f"ingested={self.ingest_count} "
                    f"phases={ {k: round(v, 1) for k, v in self.phase_seconds.items()} }")
            if max_grad_steps and self.grad_steps >= max_grad_steps:
                break
            if max_seconds and time.perf_counter() - t0 > max_seconds:
                break
        if self.device.type == "cuda":
            torch.cuda.synchronize(self.device)  # completion-accurate rate
            # flush any in-flight async publish so players/evaluators see
            # the final weights
            if self._pub_iteration and self._pub_event is not None:
                self._pub_event.synchronize()
                self.snapshot.publish(self._pub_pinned, self._pub_iteration)
                self._pub_iteration = 0
        if self.save_dir and self.grad_steps:
            save_checkpoint(self.engine, self.save_dir,
                            update_iteration=self.iteration_counter)
        wall = time.perf_counter() - t0
        return {
            "grad_steps": self.grad_steps,
            "iterations": self.iteration_counter,
            "ingested": self.ingest_count,
            "wall_seconds": round(wall, 2),
            "grad_steps_per_sec": round(self.grad_steps / max(wall, 1e-9), 1),
            "env_steps_per_sec": round(self.ingest_count / max(wall, 1e-9), 1),
            "ring_drops": sum(int(r.dropped()) for r in self.rings),
            "phase_seconds": {k: round(v, 2)
                              for k, v in self.phase_seconds.items()},
        }
