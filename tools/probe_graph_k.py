#!/usr/bin/env python3
"""Is the captured update HOST-launch-bound?  Measure (a) pure enqueue
rate of graph.replay() without sync, (b) steady rate with sync, for a
graph capturing K=1,2,4,8 updates per replay."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch  # noqa: E402

from bench import build_engine_and_replay, prefill_replay  # noqa: E402
from distributed_sac_amd.config import load_variant  # noqa: E402


def main():
    cfg = load_variant("mtsac")
    dev = "cuda:0"
    engine, replay = build_engine_and_replay(cfg, dev, "bf16", 0)

    for K in (1, 2, 4, 8):
        torch.cuda.synchronize()
        side = torch.cuda.Stream()
        side.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(side):
            for _ in range(3):
                for _ in range(K):
                    engine._update_tensors_manual(
                        replay.sample(cfg.batch_size, graph_safe=True))
        torch.cuda.current_stream().wait_stream(side)
        torch.cuda.synchronize()
        g = torch.cuda.CUDAGraph()
        with torch.cuda.graph(g):
            for _ in range(K):
                engine._update_tensors_manual(
                    replay.sample(cfg.batch_size, graph_safe=True))
        torch.cuda.synchronize()

        n = max(2000 // K, 200)
        # warm
        for _ in range(50):
            g.replay()
        torch.cuda.synchronize()
        # (a) enqueue-only
        t0 = time.perf_counter()
        for _ in range(n):
            g.replay()
        t_enq = time.perf_counter() - t0
        torch.cuda.synchronize()
        # (b) steady with end sync
        t0 = time.perf_counter()
        for _ in range(n):
            g.replay()
        torch.cuda.synchronize()
        t_tot = time.perf_counter() - t0
        ups = n * K
        print(f"K={K}: enqueue {t_enq/n*1e3:.3f} ms/replay "
              f"({t_enq/ups*1e3:.3f} ms/update) | "
              f"steady {t_tot/ups*1e3:.4f} ms/update "
              f"= {ups/t_tot:.0f} grad-steps/s")


if __name__ == "__main__":
    main()
