#!/usr/bin/env python3
"""Flagship benchmark — learner grad-steps/sec on the MT10-MTSAC config.

Driver contract:
    python bench.py --gpus N --steps K --warmup W
    (for N>1 launched via torch.distributed.run, one rank per GPU over RCCL)

Measures the BASELINE.json metric ("learner grad-steps/sec + actor
env-steps/sec, LunarLander SAC & MT10-MTSAC") on the MT10-MTSAC shipped
config: batch 1280, 39+10-d mtobs, 4-d actions, 3x400 MLPs, twin critics,
per-task alpha, weighted loss — synthetic transitions, random-init
weights.  Compute dtype defaults to bf16 (fp32 master weights, bf16 GEMM
mirrors) — the precision BASELINE.json's config list names for the MI355X
learner ("8 CPU rollout workers + 1 MI355X learner bf16"); the reference
itself is fp32-only, so the fp32 rate is measured and reported alongside
(``fp32_value``) and available via ``--dtype fp32``.  Each timed step is
ONE full SAC gradient update (TD target, critic fwd+bwd+Adam, actor
fwd+bwd+Adam, alpha fwd+bwd+Adam, Polyak) — nothing is skipped or cached.
For N>1 the learner is data-parallel: flat-bucket RCCL all-reduce of
critic/actor/alpha gradients per update (weak scaling: per-GPU batch
fixed).

Timing: the K-step window (barrier + synchronize on both sides, MAX over
ranks) is repeated until the cumulative timed region reaches
``--min-timed-seconds`` (default 2 s), and the MEDIAN window is reported —
so a small --steps can never decide the headline from a 10 ms window.
``ms_per_step`` × ``steps`` is always one real contiguous window.

`value` = aggregate grad-steps/sec = N * K / max-over-ranks(median window).
`env_steps_per_sec` (auxiliary, untimed region) = synthetic-env rollout
throughput of one vectorized player with batched policy inference.

``--async`` benches the real asynchronous topology instead (players ->
shm rings -> learner with ingest/publish on the update path), reporting
the sustained learner grad-steps/s with env-steps/s flowing concurrently.
"""

from __future__ import annotations

import argparse
import json
import os
import statistics
import sys
import time

import torch


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=300)
    p.add_argument("--warmup", type=int, default=30)
    p.add_argument("--config", type=str, default="mtsac",
                   choices=["mtsac", "sac", "vsac", "care", "mt1_care"])
    p.add_argument("--device", type=str, default=None)
    p.add_argument("--skip-rollout-probe", action="store_true")
    p.add_argument("--no-graph", action="store_true",
                   help="disable hipGraph capture of the update step")
    p.add_argument("--dtype", choices=["fp32", "bf16"], default="bf16",
                   help="GEMM compute dtype (fp32 masters either way)")
    p.add_argument("--min-timed-seconds", type=float, default=2.0,
                   help="repeat the K-step window until this much total "
                        "timed region accumulated (median window reported)")
    p.add_argument("--max-windows", type=int, default=64)
    p.add_argument("--skip-fp32-probe", action="store_true",
                   help="skip the auxiliary fp32-rate measurement")
    p.add_argument("--async", dest="async_mode", action="store_true",
                   help="bench the real async topology: players -> shm "
                        "rings -> learner (single rank)")
    p.add_argument("--async-seconds", type=float, default=30.0)
    p.add_argument("--players", type=int, default=4)
    p.add_argument("--chunk-steps", type=int, default=64)
    p.add_argument("--graph-chunk", type=int, default=4,
                   help="updates captured per hipGraph replay (single-GPU "
                        "graph mode; auto-disabled unless it divides "
                        "--steps)")
    p.add_argument("--force-ddp", action="store_true",
                   help="initialize the process group + run the DP code "
                        "path even at world_size 1 (single-GPU RCCL "
                        "validation / launch-gap profiling)")
    return p.parse_args()


def prefill_replay(replay, cfg, per_shard: int, device, seed: int):
    g = torch.Generator(device=device)
    g.manual_seed(seed)
    num_tasks = len(replay.shards)
    for t, shard in enumerate(replay.shards):
        n = per_shard
        states = torch.randn(n, cfg.mtobs_dim, device=device, generator=g)
        next_states = torch.randn(n, cfg.mtobs_dim, device=device, generator=g)
        if num_tasks > 1:
            oh = torch.zeros(n, num_tasks, device=device)
            oh[:, t] = 1.0
            states[:, -num_tasks:] = oh
            next_states[:, -num_tasks:] = oh
        shard.append(
            states,
            (torch.rand(n, cfg.action_dim, device=device, generator=g) * 2 - 1),
            torch.randn(n, 1, device=device, generator=g),
            next_states,
            (torch.rand(n, 1, device=device, generator=g) < 0.02).float(),
        )


def rollout_probe(cfg, device, envs_per_task: int = 32,
                  steps: int = 200) -> float:
    """Aux metric: env-steps/sec of ONE vectorized synthetic-env player
    (batched env stepping + batched policy inference)."""
    import copy
    from distributed_sac_amd.workers.player import build_actor
    from distributed_sac_amd.workers.rollout import FastSyntheticRollout
    c = copy.deepcopy(cfg)
    c.random_step = 0  # measure the policy-inference path, not warmup
    num_tasks = c.num_tasks if c.variant in ("mtsac", "care") else 1
    actor = build_actor(c)
    ro = FastSyntheticRollout(c, list(range(num_tasks)), actor,
                              envs_per_task=envs_per_task, seed=1234)
    ro.collect(10)  # warm
    t0 = time.perf_counter()
    ro.collect(steps)
    dt = time.perf_counter() - t0
    return steps * num_tasks * envs_per_task / dt


def build_engine_and_replay(cfg, device, precision, rank, ddp=None,
                            seed_off: int = 0):
    from distributed_sac_amd.algo import create_engine
    from distributed_sac_amd.replay import ShardedReplay
    engine = create_engine(cfg, device, precision=precision)
    if ddp is not None:
        engine.attach_ddp(ddp)
    num_tasks = cfg.num_tasks if cfg.variant in ("mtsac", "care") else 1
    replay = ShardedReplay(cfg.buffer_size, num_tasks, cfg.mtobs_dim,
                           cfg.action_dim, device=device,
                           seed=42 + rank + seed_off)
    prefill_replay(replay, cfg, per_shard=max(4096, cfg.batch_size),
                   device=device, seed=77 + rank + seed_off)
    return engine, replay


def measure(engine, replay, cfg, args, device, ddp, rank,
            label: str = "main"):
    """Warmup, then repeat the exactly-K-step timed window (barrier + sync
    brackets, max over ranks) until >= min-timed-seconds accumulated;
    return (median_window_s, n_windows, graphed)."""

    def sync():
        if device.startswith("cuda"):
            torch.cuda.synchronize()

    graphed = False
    dp_graphed = False
    # RCCL collectives are NOT capturable on this stack (the capture probe
    # aborts the process via the NCCL watchdog, not a catchable error) —
    # DP ranks use the SEGMENTED capture (three hipGraphs with the two
    # eager all-reduces between replays, engine.capture_dp).
    # chunked capture (c updates per hipGraph replay) amortizes the
    # ~0.22 ms host-side hipGraphLaunch; used only when it divides the
    # timed step count exactly so a window is EXACTLY args.steps updates
    chunk = args.graph_chunk if args.steps % args.graph_chunk == 0 else 1
    if device.startswith("cuda") and not args.no_graph:
        if ddp is None:
            try:
                engine.capture(replay, cfg.batch_size, chunk=chunk)
                graphed = True
            except Exception as e:  # pragma: no cover
                print(f"[bench] hipGraph capture failed, eager fallback: "
                      f"{e!r}", file=sys.stderr)
        elif getattr(engine, "_bf16", False):
            try:
                engine.capture_dp(replay, cfg.batch_size)
                dp_graphed = True
            except Exception as e:  # pragma: no cover
                print(f"[bench] segmented DP capture failed, eager "
                      f"fallback: {e!r}", file=sys.stderr)

    def one_step():   # one GRAD UPDATE (chunked replay runs chunk of them)
        if graphed:
            engine.graphed_update()
        elif dp_graphed:
            engine.dp_graphed_update()
        else:
            engine.update_tensors(replay.sample(cfg.batch_size,
                                                graph_safe=True))
            engine.update_iteration += 1

    n_warm = (args.warmup + chunk - 1) // chunk if graphed else args.warmup
    for _ in range(n_warm):
        one_step()

    if rank == 0 and (graphed or dp_graphed):
        print(f"[bench {label}] capture: "
              f"{'full hipGraph' if graphed else 'segmented DP hipGraphs'}",
              file=sys.stderr)
    windows = []
    while True:
        if ddp is not None:
            ddp.barrier()
        sync()
        t0 = time.perf_counter()
        for _ in range(args.steps // chunk if graphed else args.steps):
            one_step()
        sync()
        elapsed = time.perf_counter() - t0
        if ddp is not None:
            print(f"[bench {label} rank {rank}] window {len(windows)}: "
                  f"{elapsed * 1e3:.2f} ms local", file=sys.stderr)
            elapsed = ddp.max_scalar(elapsed)
        windows.append(elapsed)
        if sum(windows) >= args.min_timed_seconds \
                or len(windows) >= args.max_windows:
            break
    if ddp is not None:
        ddp.barrier()
    med = statistics.median(windows)
    if rank == 0:
        lo, hi = min(windows), max(windows)
        print(f"[bench {label}] {len(windows)} windows x {args.steps} steps: "
              f"median {med * 1e3:.2f} ms (min {lo * 1e3:.2f}, "
              f"max {hi * 1e3:.2f})", file=sys.stderr)
    mode = ("graph" if graphed
            else "dp-seg-graph" if dp_graphed else "eager")
    return med, len(windows), mode


def run_async_bench(cfg, args, device, precision):
    """--async: players -> shm rings -> learner, sustained rates."""
    from distributed_sac_amd.workers.orchestrator import DistributedTrainer
    cfg2 = cfg
    cfg2.random_step = 200
    cfg2.start_memory_len = min(cfg.start_memory_len, 2000)
    dt = DistributedTrainer(cfg2, device=device,
                            num_players=args.players,
                            chunk_steps=args.chunk_steps, seed=7,
                            use_graph=not args.no_graph,
                            precision=precision)
    stats = dt.run(max_seconds=args.async_seconds)
    return stats


def main():
    args = parse_args()
    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    # the whole-job aggregate must reflect the ranks that actually ran:
    # under the driver's torchrun launch world == --gpus; a bare
    # single-process run reports 1 even if --gpus was passed larger.
    n_gpus = world
    if args.gpus > world and rank == 0:
        print(f"[bench] note: --gpus {args.gpus} requested but "
              f"WORLD_SIZE={world}; reporting n_gpus={world}",
              file=sys.stderr)

    if args.device:
        device = args.device
    elif torch.cuda.is_available():
        device = f"cuda:{local_rank}"
        torch.cuda.set_device(device)
    else:
        device = "cpu"

    from distributed_sac_amd.config import load_variant
    from distributed_sac_amd.parallel import DataParallelGroup

    cfg = load_variant(args.config)
    cfg.device = device
    torch.manual_seed(1000 + rank)

    precision = ("bf16" if args.dtype == "bf16"
                 and device.startswith("cuda") else "fp32")

    model_desc = ("MT10-MTSAC (mtobs 49, act 4, actor 3x400, "
                  "twin critic 3x400, per-task alpha, weighted loss)"
                  if args.config == "mtsac" else args.config)

    if args.async_mode:
        assert world == 1, "--async is a single-rank bench"
        t0 = time.perf_counter()
        stats = run_async_bench(cfg, args, device, precision)
        wall = time.perf_counter() - t0
        result = {
            "metric": f"async learner grad-steps/sec (players->rings->"
                      f"learner), {args.config}",
            "value": stats.get("grad_steps_per_sec", 0.0),
            "unit": "grad_steps/s",
            "n_gpus": 1,
            "steps": stats.get("grad_steps", 0),
            "warmup": 0,
            "ms_per_step": round(1e3 / max(stats.get("grad_steps_per_sec",
                                                     1e-9), 1e-9), 4),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": precision,
            "data": "synthetic",
            "env_steps_per_sec": stats.get("env_steps_per_sec"),
            "ring_drops": stats.get("ring_drops"),
            "phase_seconds": stats.get("phase_seconds"),
            "wall_seconds": round(wall, 2),
            "config": {"model": model_desc,
                       "global_batch": cfg.batch_size,
                       "seq_len": 1,
                       "parallelism": "dp1",
                       "players": args.players,
                       "update": "full SAC step + live ingest/publish",
                       "topology": "async players->shm rings->learner"},
        }
        print(json.dumps(result))
        return

    ddp = (DataParallelGroup(device=torch.device(device),
                             force=args.force_ddp)
           if (world > 1 or args.force_ddp) else None)
    engine, replay = build_engine_and_replay(cfg, device, precision, rank,
                                             ddp=ddp)

    env_rate = None
    if rank == 0 and not args.skip_rollout_probe:
        env_rate = rollout_probe(cfg, "cpu")

    med, n_windows, graph_mode = measure(engine, replay, cfg, args, device,
                                         ddp, rank)
    steps_per_sec = args.steps / med
    value = n_gpus * steps_per_sec

    # auxiliary fp32 rate (the reference's own precision) — single-rank
    # only; reported as extra keys so the headline stays one dtype
    fp32_value = fp32_ms = None
    if precision == "bf16" and world == 1 and not args.skip_fp32_probe:
        engine32, replay32 = build_engine_and_replay(cfg, device, "fp32",
                                                     rank, seed_off=1000)
        med32, _, _ = measure(engine32, replay32, cfg, args, device,
                              None, rank, label="fp32")
        fp32_value = round(args.steps / med32, 2)
        fp32_ms = round(med32 / args.steps * 1000, 4)

    if rank == 0:
        result = {
            "metric": "learner grad-steps/sec (aggregate over GPUs), MT10-MTSAC"
                      if args.config == "mtsac" else
                      f"learner grad-steps/sec (aggregate over GPUs), {args.config}",
            "value": round(value, 2),
            "unit": "grad_steps/s",
            "n_gpus": n_gpus,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(med / args.steps * 1000, 4),
            "higher_is_better": True,
            "scaling": "weak",
            # reference publishes no grad-step rate for these exact configs
            # (MT10-MTSAC rate logs stripped); the only
            # derivable reference rate is the LunarLander config (55,290
            # grad steps over the 0.45 h logged run = ~34.1/s on a GTX
            # 1080) — reported for --config sac.
            "vs_baseline": (round(value / 34.1, 1)
                            if args.config == "sac" else None),
            "dtype": precision,
            "data": "synthetic",
            "env_steps_per_sec": (round(env_rate, 1) if env_rate else None),
            "timed_windows": n_windows,
            "fp32_value": fp32_value,
            "fp32_ms_per_step": fp32_ms,
            "config": {
                "model": model_desc,
                "global_batch": cfg.batch_size * n_gpus,
                "seq_len": 1,
                "parallelism": f"dp{n_gpus}",
                "batch_per_gpu": cfg.batch_size,
                "update": "full SAC step: TD target + critic/actor/alpha "
                          "fwd+bwd+fusedAdam + Polyak",
                # "graph" = one captured hipGraph; "dp-seg-graph" = three
                # captured segments with eager RCCL all-reduces between
                "hipgraph": graph_mode != "eager",
                "launch_mode": graph_mode,
                "graph_chunk": (args.graph_chunk
                                if graph_mode == "graph"
                                and args.steps % args.graph_chunk == 0
                                else 1),
            },
        }
        print(json.dumps(result))


if __name__ == "__main__":
    main()
