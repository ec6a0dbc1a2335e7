import sys, torch
sys.path.insert(0, "/root/repo")
sys.path.insert(0, "/root/repo/tools")
from validate_chain_fwd import per_layer_ref, ext, dev
torch.manual_seed(0)
M, dims, G = 2560, [49,400,400,400,8], 1
x1 = torch.randn(M, dims[0], device=dev); x2e = torch.empty(0, device=dev)
ws, bs, K = [], [], dims[0]
for N in dims[1:]:
    ws.append((torch.randn(N, K, device=dev)/K**0.5).to(torch.bfloat16).contiguous())
    bs.append(torch.randn(N, device=dev)); K = N
for _ in range(20): ext.mlp_chain_fwd_bf16(x1, x2e, ws, bs, 0, 1, 1, 1)
torch.cuda.synchronize()
for _ in range(300): ext.mlp_chain_fwd_bf16(x1, x2e, ws, bs, 0, 1, 1, 1)
torch.cuda.synchronize()
