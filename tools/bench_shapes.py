#!/usr/bin/env python3
"""Per-shape times for the three bf16 GEMM kernels at the REAL layer
shapes of the MTSAC update (the profile CSV only shows per-kernel-name
aggregates).  Run on a GPU box: python tools/bench_shapes.py
"""

import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from distributed_sac_amd import ops

ext = ops.native()
dev = "cuda:0"


def timeit(fn, iters=300):
    for _ in range(10):
        fn()
    torch.cuda.synchronize()
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


def bf(*shape):
    return torch.randn(*shape, device=dev).to(torch.bfloat16).contiguous()


def main():
    # (label, M, K, N, G) — per-launch shapes in one MTSAC update
    fwd_shapes = [
        ("actor_L0   ", 2560, 49, 400, 1),
        ("actor_L1   ", 2560, 400, 400, 1),
        ("actor_head ", 2560, 400, 8, 1),
        ("twin_L0    ", 1280, 53, 400, 2),
        ("twin_L1    ", 1280, 400, 400, 2),
        ("twin_head  ", 1280, 400, 1, 2),
    ]
    print("== k_bf16_fwd (per launch) ==")
    tot = 0.0
    # per update: actor chain once (4), target/critic/actor-side twins (12)
    counts = {"actor_L0   ": 1, "actor_L1   ": 2, "actor_head ": 1,
              "twin_L0    ": 3, "twin_L1    ": 6, "twin_head  ": 3}
    for lbl, M, K, N, G in fwd_shapes:
        x = bf(M, K) if G == 1 else bf(M, K)
        w = bf(G, N, K) if G > 1 else bf(N, K)
        b = torch.randn(G, N, device=dev) if G > 1 else torch.randn(N, device=dev)
        us = timeit(lambda: ext.linear_act_fwd_bf16(x, w, b, 1, G, 0))
        flops = 2.0 * M * K * N * G
        print(f"  {lbl} M{M:5d} K{K:4d} N{N:4d} G{G}  {us:7.2f} us  "
              f"{flops/us/1e6:7.1f} GFLOP/s  x{counts[lbl]}")
        tot += us * counts[lbl]
    print(f"  fwd total per update ~ {tot:.1f} us")

    print("== k_bf16_dx (per launch) ==")
    dx_shapes = [
        ("actor_head ", 2560, 8, 400, 1, 0),
        ("actor_L1   ", 2560, 400, 400, 1, 0),
        ("twin_head  ", 1280, 1, 400, 2, 0),
        ("twin_L1    ", 1280, 400, 400, 2, 0),
        ("twin_L0sum ", 1280, 400, 53, 2, 1),
    ]
    counts_dx = {"actor_head ": 1, "actor_L1   ": 2, "twin_head  ": 2,
                 "twin_L1    ": 4, "twin_L0sum ": 1}
    tot = 0.0
    for lbl, M, N, K, G, sumg in dx_shapes:
        dy = bf(M, N) if G == 1 else bf(G, M, N)
        w = bf(G, N, K) if G > 1 else bf(N, K)
        yo = dy  # relu mask tensor is dy-shaped; act=0 leaves it unused
        us = timeit(lambda: ext.linear_bwd_dx_bf16(dy, w, yo, 0, G, sumg))
        flops = 2.0 * M * K * N * G
        print(f"  {lbl} M{M:5d} N{N:4d} K{K:4d} G{G}  {us:7.2f} us  "
              f"{flops/us/1e6:7.1f} GFLOP/s  x{counts_dx[lbl]}")
        tot += us * counts_dx[lbl]
    print(f"  dx total per update ~ {tot:.1f} us")

    print("== k_bf16_dwdb_splitk (per launch, arena variant not used here) ==")
    dw_shapes = [
        ("actor_head ", 2560, 8, 400, 1),
        ("actor_L1   ", 2560, 400, 400, 1),
        ("actor_L0   ", 2560, 400, 49, 1),
        ("twin_head  ", 1280, 1, 400, 2),
        ("twin_L1    ", 1280, 400, 400, 2),
        ("twin_L0    ", 1280, 400, 53, 2),
    ]
    tot = 0.0
    counts_dw = {"actor_head ": 1, "actor_L1   ": 2, "actor_L0   ": 1,
                 "twin_head  ": 1, "twin_L1    ": 2, "twin_L0    ": 1}
    for lbl, M, N, K, G in dw_shapes:
        dy = bf(M, N) if G == 1 else bf(G, M, N)
        x = bf(M, K)
        yo = dy
        us = timeit(lambda: ext.linear_bwd_dwdb_bf16(dy, x, yo, 0, G))
        flops = 2.0 * M * K * N * G
        print(f"  {lbl} M{M:5d} N{N:4d} K{K:4d} G{G}  {us:7.2f} us  "
              f"{flops/us/1e6:7.1f} GFLOP/s  x{counts_dw[lbl]}")
        tot += us * counts_dw[lbl]
    print(f"  dwdb total per update ~ {tot:.1f} us (critic path uses arena variant)")


if __name__ == "__main__":
    main()
