import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from distributed_sac_amd.config import load_variant
from distributed_sac_amd.workers.learner import Learner
from distributed_sac_amd.workers.param_server import ParamSnapshot
from distributed_sac_amd.workers.orchestrator import _actor_numel
import queue

cfg = load_variant("mtsac")
snap = ParamSnapshot(_actor_numel(cfg))
lr = Learner(cfg, "cuda:0", snap, queue.Queue(), use_graph=True)
# prefill
import bench
bench.prefill_replay(lr.replay, cfg, 4096, "cuda:0", 7)
lr.engine.hard_copy_targets()

# timed publish sub-steps
flat = lr.engine.publish_params()
torch.cuda.synchronize()
t0 = time.perf_counter()
for _ in range(50):
    lr.publish()
print("publish avg ms (no players):", (time.perf_counter() - t0) / 50 * 1e3)

pinned = torch.empty(flat.numel(), pin_memory=True)
torch.cuda.synchronize()
t0 = time.perf_counter()
for _ in range(50):
    pinned.copy_(flat.reshape(-1))
print("d2h pinned avg ms:", (time.perf_counter() - t0) / 50 * 1e3)
t0 = time.perf_counter()
for _ in range(50):
    snap.buf.copy_(pinned)
print("shm memcpy avg ms:", (time.perf_counter() - t0) / 50 * 1e3)
t0 = time.perf_counter()
for _ in range(50):
    snap.meta[0] = int(snap.meta[0]) + 1
print("meta scalar avg ms:", (time.perf_counter() - t0) / 50 * 1e3)

# the full learner loop rate without players
t0 = time.perf_counter()
stats = lr.run(max_grad_steps=300, max_seconds=60)
print("no-player stats:", stats)
