"""distributed_sac_amd — MI355X-native distributed Soft Actor-Critic framework.

A from-scratch re-design of the capabilities of SKSKSK94/Distributed_SAC
(vanilla SAC, MT-SAC, CARE on LunarLander / Meta-World MT1 / MT10) for AMD
Instinct MI355X (gfx950):

- SAC hot-path math runs in hand-written CDNA4 HIP kernels (MFMA GEMMs with
  fused bias+activation, fused tanh-Gaussian sampling + log-prob, TD-target /
  loss reductions, multi-tensor Adam, Polyak updates) — see
  ``distributed_sac_amd.ops``.
- The replay buffer is GPU-resident in HBM3E (the reference's 1e6-transition
  deque is ~250 MB — trivial against 288 GB), sampled by device-side gather;
  rollout transitions arrive through pinned-host staging, replacing the
  reference's Redis+pickle data plane (reference replay_buffer.py).
- Data-parallel learners scale over the 8 GPUs of a node with one process per
  GPU and a single flat-bucket RCCL all-reduce per update over xGMI
  (``distributed_sac_amd.parallel``) — net-new vs the single-GPU reference.
- Config JSON and checkpoint ``.tar`` formats are reference-compatible
  (``distributed_sac_amd.config`` / ``distributed_sac_amd.checkpoint``).

Pure-torch fp32 reference implementations of every op live in
``ops.torch_ref``; they are the CPU execution path and the numerics oracle
for the HIP kernels.
"""

__version__ = "0.1.0"

from . import config  # noqa: F401
