import os, sys, time
sys.path.insert(0, "/root/repo")
import torch
from distributed_sac_amd import ops
ext = ops.native()
B, H, K = 1280, 400, 400
x = torch.randn(B, H, device="cuda")
w = torch.randn(2, H, H, device="cuda") / H ** 0.5
b2 = torch.randn(2, H, device="cuda")
xh = x.to(torch.bfloat16)
wh = w.to(torch.bfloat16)
for _ in range(20):
    y32 = ext.linear_act_fwd_g(x, w, b2, 1, 2)
    y16 = ext.linear_act_fwd_bf16(xh, wh, b2, 1, 2, 0)
torch.cuda.synchronize()
for _ in range(50):
    y32 = ext.linear_act_fwd_g(x, w, b2, 1, 2)
torch.cuda.synchronize()
for _ in range(50):
    y16 = ext.linear_act_fwd_bf16(xh, wh, b2, 1, 2, 0)
torch.cuda.synchronize()
print("done")
