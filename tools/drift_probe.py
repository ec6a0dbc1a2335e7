#!/usr/bin/env python3
"""CPU-learner long-run drift investigation (round-1 observation:
33->45 ms/update over 2 h with flat RSS in the async CPU soak).

Runs the async CPU topology for --hours, logging the learner rate every
30 s and cProfile-ing a 200-update window every 30 minutes; windows are
compared at the end to show WHERE the extra time went.
Writes a report to profiles/r17_cpu_drift.md.
"""

import cProfile
import io
import os
import pstats
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from distributed_sac_amd.workers.orchestrator import DistributedTrainer
from tests.test_trainer import tiny_cfg


def main():
    hours = float(sys.argv[1]) if len(sys.argv) > 1 else 2.0
    torch.manual_seed(0)
    cfg = tiny_cfg("mtsac")
    cfg.actor_hidden_dim = cfg.critic_hidden_dim = [64, 64]
    cfg.batch_size = 256
    cfg.buffer_size = 200_000
    cfg.start_memory_len = 500
    cfg.random_step = 200
    cfg.update_delay = 1
    dt = DistributedTrainer(cfg, device="cpu", num_players=2,
                            chunk_steps=64, seed=3, use_graph=False)
    dt.start_players()
    lr = dt.learner
    lr.publish()
    t0 = time.perf_counter()
    while not lr.ready():
        lr.drain_queue()
        time.sleep(0.01)
    lr.engine.hard_copy_targets()
    print("start train", flush=True)

    rates = []       # (elapsed_min, updates_per_s)
    prof_dumps = []  # (elapsed_min, text)
    last_rate_t = time.perf_counter()
    last_steps = 0
    next_prof = 0.0
    deadline = t0 + hours * 3600

    while time.perf_counter() < deadline:
        now = time.perf_counter()
        el_min = (now - t0) / 60
        if el_min >= next_prof:
            pr = cProfile.Profile()
            pr.enable()
            for _ in range(200):
                lr.drain_queue()
                lr.train_step()
            pr.disable()
            s = io.StringIO()
            pstats.Stats(pr, stream=s).sort_stats("cumulative").print_stats(18)
            prof_dumps.append((round(el_min, 1), s.getvalue()))
            next_prof = el_min + 30.0
            print(f"[{el_min:.1f} min] profiled window", flush=True)
        for _ in range(50):
            lr.drain_queue()
            lr.train_step()
        now = time.perf_counter()
        if now - last_rate_t >= 30.0:
            d = lr.grad_steps - last_steps
            rates.append((round((now - t0) / 60, 1),
                          round(d / (now - last_rate_t), 1)))
            last_steps = lr.grad_steps
            last_rate_t = now
            print(f"[{rates[-1][0]} min] {rates[-1][1]} upd/s", flush=True)
    dt.shutdown()

    import resource
    rss = resource.getrusage(resource.RUSAGE_SELF).ru_maxrss // 1024
    with open("profiles/r17_cpu_drift.md", "w") as f:
        f.write("# r17 — CPU-learner drift probe (%.1f h, tiny MTSAC, "
                "2 players)\n\nmax RSS %d MB\n\n## rate curve "
                "(min, updates/s)\n\n" % (hours, rss))
        for t, r in rates:
            f.write(f"- {t} min: {r}/s\n")
        f.write("\n## cProfile windows (200 updates each)\n")
        for t, txt in prof_dumps:
            f.write(f"\n### at {t} min\n```\n")
            f.write("\n".join(txt.splitlines()[:28]))
            f.write("\n```\n")
    print("report written", flush=True)


if __name__ == "__main__":
    main()
