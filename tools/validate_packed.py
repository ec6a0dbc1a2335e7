#!/usr/bin/env python3
"""GPU validation + A/B for the FRAGMENT-PACKED chain paths.
Run on a GPU box: python tools/validate_packed.py"""

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


def packed_of(w, G, dx):
    K = w.shape[-1]
    N = w.numel() // (G * K)
    rows, cols = (K, N) if dx else (N, K)
    NT, KS = (rows + 15) // 16, (cols + 31) // 32
    p = torch.empty(G * NT * KS * 512, device=dev, dtype=torch.bfloat16)
    ext.pack_weights_frag([w], [p], [G], [1 if dx else 0])
    return p


def bf(*shape):
    return (torch.randn(*shape, device=dev) / 3).to(torch.bfloat16).contiguous()


def run_fwd(name, M, dims, G, two_src=True, bench=False):
    torch.manual_seed(0)
    K0 = dims[0]
    if two_src:
        x1 = torch.randn(M, K0 - 4, device=dev)
        x2 = torch.randn(M, 4, device=dev)
    else:
        x1 = torch.randn(M, K0, device=dev)
        x2 = torch.empty(0, device=dev)
    ws, bs, K = [], [], K0
    for N in dims[1:]:
        w = bf(G, N, K) if G > 1 else bf(N, K)
        ws.append(w)
        bs.append(torch.randn(G, N, device=dev) if G > 1
                  else torch.randn(N, device=dev))
        K = N
    wps = [packed_of(w, G, False) for w in ws]
    ref = ext.mlp_chain_fwd_bf16(x1, x2, ws, bs, 0, G, 1, 0, 0, 1)
    out = ext.mlp_chain_fwd_bf16(x1, x2, ws, bs, 0, G, 1, 0, 0, 1, wps)
    d = max((a.float() - b.float()).abs().max().item()
            for a, b in zip(out, ref))
    check(f"fwd {name} packed==unpacked (d={d})", d == 0)
    if bench:
        t0 = timeit(lambda: ext.mlp_chain_fwd_bf16(
            x1, x2, ws, bs, 0, G, 1, 0, 0, 1))
        t1 = timeit(lambda: ext.mlp_chain_fwd_bf16(
            x1, x2, ws, bs, 0, G, 1, 0, 0, 1, wps))
        tp = timeit(lambda: ext.pack_weights_frag(
            ws, wps, [G] * len(ws), [0] * len(ws)))
        print(f"      unpacked {t0:8.2f} us  packed {t1:8.2f} us "
              f"({t0 / t1:.2f}x)  [pack launch {tp:.2f} us]")


def run_dx(name, M, dims, G, bench=False):
    torch.manual_seed(1)
    K0 = dims[0]
    ws, wts, wps, youts, flags, K = [], [], [], [], [], K0
    L = len(dims) - 1
    for i, N in enumerate(dims[1:]):
        last = i == L - 1
        w = bf(G, N, K)
        ws.append(w)
        wt = torch.empty(G, K, N, device=dev, dtype=torch.bfloat16)
        ext.transpose_weights_bf16([w], [wt], [G])
        wts.append(wt)
        wps.append(packed_of(w, G, True))
        flags.append(0 if last else 1)
        ysh = (G, M, N) if G > 1 else (M, N)
        youts.append(torch.empty(0, device=dev, dtype=torch.bfloat16)
                     if last else
                     (torch.randn(*ysh, device=dev).relu())
                     .to(torch.bfloat16).contiguous())
        K = N
    dy = bf(G, M, dims[-1]) if G > 1 else bf(M, dims[-1])
    ref = ext.mlp_chain_dx_bf16(dy, wts, youts, K0, flags, G, 1, 2)
    out = ext.mlp_chain_dx_bf16(dy, wts, youts, K0, flags, G, 1, 2, wps)
    d = max((a.float() - b.float()).abs().max().item()
            for a, b in zip(out, ref))
    check(f"dx {name} packed==unpacked (d={d})", d == 0)
    if bench:
        t0 = timeit(lambda: ext.mlp_chain_dx_bf16(
            dy, wts, youts, K0, flags, G, 1, 2))
        t1 = timeit(lambda: ext.mlp_chain_dx_bf16(
            dy, wts, youts, K0, flags, G, 1, 2, wps))
        print(f"      unpacked {t0:8.2f} us  packed {t1:8.2f} us "
              f"({t0 / t1:.2f}x)")


def main():
    assert torch.cuda.is_available()
    # pack correctness vs torch
    torch.manual_seed(2)
    w = bf(2, 37, 53)
    p = packed_of(w, 2, False)
    NT, KS = (37 + 15) // 16, (53 + 31) // 32
    pv = p.view(2, NT, KS, 64, 8)
    okp = True
    for g in (0, 1):
        for t in (0, NT - 1):
            for ks in (0, KS - 1):
                for lane in (0, 17, 63):
                    row = t * 16 + (lane & 15)
                    k0 = ks * 32 + (lane >> 4) * 8
                    for j in (0, 7):
                        exp = (w[g, row, k0 + j].item()
                               if row < 37 and k0 + j < 53 else 0.0)
                        okp &= float(pv[g, t, ks, lane, j]) == exp
    check("pack fwd layout spot-check", okp)
    wp = packed_of(w, 2, True)
    NT2, KS2 = (53 + 15) // 16, (37 + 31) // 32
    pv2 = wp.view(2, NT2, KS2, 64, 8)
    okp = True
    for lane in (0, 33):
        krow = 0 * 16 + (lane & 15)
        n0 = 0 * 32 + (lane >> 4) * 8
        for j in (0, 5):
            exp = (w[0, n0 + j, krow].item()
                   if krow < 53 and n0 + j < 37 else 0.0)
            okp &= float(pv2[0, 0, 0, lane, j]) == exp
    check("pack dx layout spot-check", okp)

    run_fwd("actor mtsac", 2560, [49, 400, 400, 400, 8], 1, two_src=False,
            bench=True)
    run_fwd("twin mtsac ", 1280, [53, 400, 400, 400, 1], 2, bench=True)
    run_fwd("odd/768    ", 512, [768, 100, 50], 1, two_src=False)
    run_dx("twin care  ", 1280, [104, 400, 400, 1], 2, bench=True)
    run_dx("actor      ", 1280, [49, 400, 400, 8], 1)
    print("PASS" if ok_all else "FAIL")
    return 0 if ok_all else 1


if __name__ == "__main__":
    sys.exit(main())
