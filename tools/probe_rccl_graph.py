"""Probe: is a torch.distributed (RCCL) all-reduce capturable in a hipGraph
on this stack? world_size=1 exercises the same capture path."""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
os.environ.setdefault("MASTER_PORT", "29871")
os.environ.setdefault("WORLD_SIZE", "1")
os.environ.setdefault("RANK", "0")
import torch
import torch.distributed as dist
dist.init_process_group("nccl")
x = torch.ones(1_000_000, device="cuda:0")
dist.all_reduce(x)  # warm
torch.cuda.synchronize()
g = torch.cuda.CUDAGraph()
s = torch.cuda.Stream()
s.wait_stream(torch.cuda.current_stream())
with torch.cuda.stream(s):
    dist.all_reduce(x)
torch.cuda.current_stream().wait_stream(s)
try:
    with torch.cuda.graph(g):
        dist.all_reduce(x)
        x.mul_(0.5)
    for _ in range(5):
        g.replay()
    torch.cuda.synchronize()
    print("RCCL graph capture: OK, x[0] =", float(x[0]))
except Exception as e:
    print("RCCL graph capture FAILED:", repr(e))
dist.destroy_process_group()
