#!/usr/bin/env python3
"""Kernel microbenchmarks: isolated times for the hot kernels at MTSAC
shapes + the per-kernel dispatch floor inside hipGraph replay.

Usage (on a GPU box):  python tools/bench_kernels.py
"""

import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from distributed_sac_amd import ops

ext = ops.native()
dev = "cuda:0"


def timeit(fn, iters=300, graph=True):
    for _ in range(10):
        fn()
    torch.cuda.synchronize()
    if graph:
        g = torch.cuda.CUDAGraph()
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            fn()
        torch.cuda.current_stream().wait_stream(s)
        with torch.cuda.graph(g):
            fn()
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(iters):
            g.replay()
        torch.cuda.synchronize()
        return (time.perf_counter() - t0) / iters * 1e6
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6


def main():
    B, H, A = 1280, 400, 4
    Din, Dc = 49, 54
    x = torch.randn(B, H, device=dev)
    x0 = torch.randn(B, Dc, device=dev)
    w = torch.randn(2, H, H, device=dev) / H ** 0.5
    w0 = torch.randn(2, H, Dc, device=dev) / Dc ** 0.5
    b2 = torch.randn(2, H, device=dev)
    wh = torch.randn(2, 1, H, device=dev)
    bh = torch.randn(2, 1, device=dev)
    dy = torch.randn(2, B, H, device=dev)
    y2 = torch.relu(torch.randn(2, B, H, device=dev))
    flat = torch.randn(2_000_000, device=dev)
    flat2 = torch.randn(2_000_000, device=dev)

    rows = []
    rows.append(("fwd G=2 1280x400x400 relu",
                 timeit(lambda: ext.linear_act_fwd_g(x, w, b2, 1, 2))))
    rows.append(("fwd G=2 1280x400x54 relu",
                 timeit(lambda: ext.linear_act_fwd_g(x0, w0, b2, 1, 2))))
    rows.append(("fwd G=2 1280x1x400 head",
                 timeit(lambda: ext.linear_act_fwd_g(x, wh, bh, 0, 2))))
    rows.append(("dx  G=2 per-group 400->400",
                 timeit(lambda: ext.linear_bwd_dx_g(dy, w, y2, 1, 2, 0))))
    rows.append(("dx  G=2 summed 400->54",
                 timeit(lambda: ext.linear_bwd_dx_g(dy, w0, y2, 1, 2, 1))))
    rows.append(("dwdb G=2 400x400 (M=1280)",
                 timeit(lambda: ext.linear_bwd_dwdb_g(dy, x, y2, 1, 2))))
    rows.append(("dwdb G=2 400x54",
                 timeit(lambda: ext.linear_bwd_dwdb_g(dy, x0, y2, 1, 2))))
    rows.append(("polyak 2M", timeit(lambda: ext.polyak_(flat, flat2, 0.005))))
    tiny = torch.zeros(8, device=dev)
    tiny2 = torch.zeros(8, device=dev)
    rows.append(("tiny polyak x1 (dispatch floor)",
                 timeit(lambda: ext.polyak_(tiny, tiny2, 0.5))))

    def fifty():
        for _ in range(50):
            ext.polyak_(tiny, tiny2, 0.5)
    rows.append(("tiny polyak x50 / 50",
                 timeit(fifty) / 50))

    for name, us in rows:
        print(f"{name:38s} {us:9.2f} us")


if __name__ == "__main__":
    main()
