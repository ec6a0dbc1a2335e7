#!/usr/bin/env python3
"""GPU validation + A/B for the fused wide MLP-chain forward
(k_bf16_chain_fwd) against the per-layer k_bf16_fwd path.

Run on a GPU box: python tools/validate_chain_fwd.py
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


def per_layer_ref(x1, x2, ws, bs, act_last, G, out_f32):
    """Existing path: cat + cast + per-layer k_bf16_fwd."""
    x = torch.cat([x1, x2], dim=-1) if x2 is not None else x1
    xh = x.to(torch.bfloat16)
    acts = [xh]
    h = xh
    L = len(ws)
    for i in range(L):
        last = i == L - 1
        h = ext.linear_act_fwd_bf16(h, ws[i], bs[i],
                                    act_last if last else 1, G,
                                    out_f32 if last else 0)
        acts.append(h)
    return h, acts


def run_case(name, M, dims, G, two_src=False, act_last=0, out_f32=1,
             bench=False, in_bf16=False, rm=0):
    torch.manual_seed(0)
    K0 = dims[0]
    if two_src:
        C1 = K0 - 4
        x1 = torch.randn(M, C1, device=dev)
        x2 = torch.randn(M, 4, device=dev)
    else:
        x1 = torch.randn(M, K0, device=dev)
        x2 = None
    if in_bf16:
        x1 = x1.to(torch.bfloat16)
        x2 = x2.to(torch.bfloat16) if x2 is not None else None
    ws, bs = [], []
    K = K0
    for N in dims[1:]:
        w = (torch.randn(G, N, K, device=dev) / K ** 0.5).to(torch.bfloat16)
        if G == 1:
            w = w[0]
        ws.append(w.contiguous())
        bs.append(torch.randn(G, N, device=dev).squeeze(0).contiguous()
                  if G == 1 else torch.randn(G, N, device=dev))
        K = N
    y_ref, acts_ref = per_layer_ref(x1, x2, ws, bs, act_last, G, out_f32)
    out = ext.mlp_chain_fwd_bf16(
        x1, x2 if x2 is not None else torch.empty(0, device=dev),
        ws, bs, act_last, G, out_f32, rm)
    y, xsave = out[0], out[1]
    ok = True
    d0 = (xsave.float() - acts_ref[0].float()).abs().max().item()
    ok &= d0 == 0
    dy = (y.float() - y_ref.float()).abs().max().item()
    ref_scale = y_ref.float().abs().max().item() + 1e-6
    ok &= dy <= 2e-2 * ref_scale + 1e-4
    dacts = []
    for i, a in enumerate(out[2:]):
        da = (a.float() - acts_ref[i + 1].float()).abs().max().item()
        dacts.append(da)
        ok &= da <= 2e-2 * acts_ref[i + 1].float().abs().max().item() + 1e-4
    print(f"  {name}: xsave_d={d0:.1e} y_d={dy:.3e} acts_d="
          f"{[f'{v:.1e}' for v in dacts]} -> {'OK' if ok else 'FAIL'}")
    if bench:
        x2e = x2 if x2 is not None else torch.empty(0, device=dev)
        t_ref = timeit(lambda: per_layer_ref(x1, x2, ws, bs, act_last, G,
                                             out_f32))
        t1 = timeit(lambda: ext.mlp_chain_fwd_bf16(
            x1, x2e, ws, bs, act_last, G, out_f32, 1))
        t2 = timeit(lambda: ext.mlp_chain_fwd_bf16(
            x1, x2e, ws, bs, act_last, G, out_f32, 2))
        print(f"      per-layer {t_ref:8.2f} us   chain rm1 {t1:8.2f} us "
              f"({t_ref / t1:.2f}x)   rm2 {t2:8.2f} us ({t_ref / t2:.2f}x)")
    return ok


def main():
    assert torch.cuda.is_available()
    ok = True
    print("== correctness ==")
    ok &= run_case("tiny           ", 64, [20, 32, 16], 1)
    ok &= run_case("odd-K0         ", 200, [49, 64, 8], 1)
    ok &= run_case("actor mtsac    ", 2560, [49, 400, 400, 400, 8], 1,
                   act_last=0, out_f32=1, bench=True)
    ok &= run_case("twin mtsac     ", 1280, [53, 400, 400, 400, 1], 2,
                   two_src=True, act_last=0, out_f32=1, bench=True)
    ok &= run_case("twin ll        ", 256, [10, 256, 256, 1], 2,
                   two_src=True, bench=True)
    ok &= run_case("bf16-in        ", 512, [104, 400, 1], 2, in_bf16=True)
    ok &= run_case("width512       ", 512, [512, 512, 16], 1)
    ok &= run_case("rm1-forced     ", 2560, [49, 400, 400, 8], 1, rm=1)
    ok &= run_case("rm2-forced     ", 1280, [53, 400, 400, 1], 2,
                   two_src=True, rm=2)
    ok &= run_case("rm2-oddM       ", 1304, [53, 400, 400, 1], 2,
                   two_src=True, rm=2)
    print("PASS" if ok else "FAIL")
    return 0 if ok else 1


if __name__ == "__main__":
    sys.exit(main())
