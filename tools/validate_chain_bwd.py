#!/usr/bin/env python3
"""GPU validation for the fused backward pieces:
  - transpose_weights_bf16 (multi-tensor W -> W^T mirrors)
  - mlp_chain_dx_bf16 (fused dx chain: mask + save + propagate [+ dx0])
  - dwdb_grouped_arena (one-launch multi-layer dW/db split-K partials)

Run on a GPU box: python tools/validate_chain_bwd.py
"""

import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from distributed_sac_amd import ops

ext = ops.native()
dev = "cuda:0"
ok_all = True


def check(name, cond):
    global ok_all
    ok_all &= bool(cond)
    print(f"  {name}: {'OK' if cond else 'FAIL'}")


def timeit(fn, iters=200):
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
    return (torch.randn(*shape, device=dev) / 3).to(torch.bfloat16).contiguous()


def test_transpose():
    torch.manual_seed(0)
    # G=1, 2-D views
    ws2 = [bf(400, 53), bf(400, 400), bf(1, 400), bf(8, 400)]
    wt2 = [torch.empty(w.shape[1], w.shape[0], device=dev,
                       dtype=torch.bfloat16) for w in ws2]
    ext.transpose_weights_bf16(ws2, wt2, [1] * len(ws2))
    d = max((a.t().contiguous() - b).abs().max().item()
            for a, b in zip(ws2, wt2))
    check(f"transpose 2-D (d={d})", d == 0)
    # G=2, 3-D
    ws3 = [bf(2, 400, 53), bf(2, 1, 400), bf(2, 400, 400)]
    wt3 = [torch.empty(w.shape[0], w.shape[2], w.shape[1], device=dev,
                       dtype=torch.bfloat16) for w in ws3]
    ext.transpose_weights_bf16(ws3, wt3, [2] * len(ws3))
    d = max((a.transpose(1, 2).contiguous() - b).abs().max().item()
            for a, b in zip(ws3, wt3))
    check(f"transpose G=2 (d={d})", d == 0)


def test_chain_dx():
    torch.manual_seed(1)
    G, M = 2, 512
    dims = [104, 400, 400, 1]  # K0, widths...
    K0 = dims[0]
    ws, wts, youts, acts_flags = [], [], [], []
    K = K0
    for i, N in enumerate(dims[1:]):
        L = len(dims) - 1
        w = bf(G, N, K)
        ws.append(w)
        wt = torch.empty(G, K, N, device=dev, dtype=torch.bfloat16)
        ext.transpose_weights_bf16([w], [wt], [G])
        wts.append(wt)
        last = i == L - 1
        acts_flags.append(0 if last else 1)
        youts.append(torch.empty(0, device=dev, dtype=torch.bfloat16)
                     if last else (bf(G, M, N).abs() *
                                   (torch.rand(G, M, N, device=dev) > 0.3)
                                   ).to(torch.bfloat16).contiguous())
        K = N

    dy_last = bf(G, M, dims[-1])
    out = ext.mlp_chain_dx_bf16(dy_last, wts, youts, K0, acts_flags, G,
                                1, 100)
    dys, dx0 = out[:-1], out[-1]

    # torch reference
    dyp = dy_last.float()
    ref_dys = [None] * len(ws)
    for l in range(len(ws) - 1, -1, -1):
        if acts_flags[l]:
            dyp = dyp * (youts[l].float() != 0)
        ref_dys[l] = dyp.to(torch.bfloat16)
        dyp = torch.einsum("gmn,gnk->gmk", ref_dys[l].float(),
                           ws[l].float())
    ref_dx0 = dyp[..., 100:]

    d = max((a.float() - b.float()).abs().max().item()
            for a, b in zip(dys, ref_dys))
    check(f"chain_dx masked dys (d={d:.2e})", d <= 2e-2)
    dd = (dx0 - ref_dx0).abs().max().item()
    sc = ref_dx0.abs().max().item()
    check(f"chain_dx dx0 cols>=100 (d={dd:.2e} scale {sc:.1f})",
          dd <= 1e-2 * sc + 1e-3)

    # A/B vs per-layer dx path
    def per_layer():
        dy = dy_last
        for l in range(len(ws) - 1, 0, -1):
            yo = youts[l] if acts_flags[l] else dys[l]
            dy = ext.linear_bwd_dx_bf16(dy, ws[l], yo, acts_flags[l], G, 0)
        return dy
    t_ref = timeit(per_layer)
    t_new = timeit(lambda: ext.mlp_chain_dx_bf16(
        dy_last, wts, youts, K0, acts_flags, G, 1, 100))
    print(f"      per-layer dx {t_ref:8.2f} us   chain_dx {t_new:8.2f} us "
          f"({t_ref / t_new:.2f}x)  [chain includes masks+saves+dx0]")


def test_grouped_dwdb():
    torch.manual_seed(2)
    G, M, S = 2, 1280, 8
    chunk = (M + S - 1) // S
    chunk = (chunk + 63) // 64 * 64
    layers = [(400, 53), (400, 400), (1, 400)]  # (N, K)
    dys = [bf(G, M, N) for N, K in layers]
    xs = [bf(M, K) if i == 0 else bf(G, M, K)
          for i, (N, K) in enumerate(layers)]
    numel = sum(G * (N * K + N) for N, K in layers)
    offs, w_offs, b_offs = 0, [], []
    for N, K in layers:
        w_offs.append(offs)
        offs += G * N * K
        b_offs.append(offs)
        offs += G * N
    arena1 = torch.zeros(S, numel, device=dev)
    arena2 = torch.zeros(S, numel, device=dev)
    out1 = torch.zeros(numel, device=dev)
    out2 = torch.zeros(numel, device=dev)

    ext.dwdb_grouped_arena(dys, xs, arena1, w_offs, b_offs, G, S, chunk)
    ext.reduce_arena(arena1, out1, S, 0, -1)

    for i, (N, K) in enumerate(layers):
        ext.linear_bwd_dwdb_arena(dys[i], xs[i], dys[i], 0, G, arena2,
                                  w_offs[i], b_offs[i], S, chunk, 0)
    ext.reduce_arena(arena2, out2, S, 0, -1)
    d = (out1 - out2).abs().max().item()
    check(f"grouped dwdb == per-layer arena (d={d})", d == 0)

    t_ref = timeit(lambda: [ext.linear_bwd_dwdb_arena(
        dys[i], xs[i], dys[i], 0, G, arena2, w_offs[i], b_offs[i], S,
        chunk, 0) for i in range(len(layers))])
    t_new = timeit(lambda: ext.dwdb_grouped_arena(
        dys, xs, arena1, w_offs, b_offs, G, S, chunk))
    print(f"      per-layer dwdb {t_ref:8.2f} us   grouped {t_new:8.2f} us "
          f"({t_ref / t_new:.2f}x)")


def main():
    assert torch.cuda.is_available()
    print("== transpose ==")
    test_transpose()
    print("== chain_dx ==")
    test_chain_dx()
    print("== grouped dwdb ==")
    test_grouped_dwdb()
    print("PASS" if ok_all else "FAIL")
    return 0 if ok_all else 1


if __name__ == "__main__":
    sys.exit(main())
