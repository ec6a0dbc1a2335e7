"""Versioned actor-weight snapshot over shared memory.

Replaces the reference's weight broadcast-by-Redis-key-overwrite
(C3/C4 in SURVEY §2.7: learner ``set('parameters', pickle(state_dict))``,
players ``get`` + unpickle EVERY env step).  Here the learner publishes its
flat fp32 actor buffer into a shared-memory tensor guarded by a seqlock;
players poll only the 16-byte version word per env step and copy the
buffer ONLY when the iteration advanced — keeping the reference's
"apply only when update_iteration changed" semantics (player.pull_parameters,
LunarLander…/src/player.py:75-85) without serialization.
"""

from __future__ import annotations

from typing import Optional

import torch


class ParamSnapshot:
    """Single-writer / multi-reader seqlock snapshot of a flat buffer."""

    def __init__(self, numel: int):
        self.buf = torch.zeros(numel, dtype=torch.float32)
        # [0]=sequence (even = stable), [1]=update_iteration
        self.meta = torch.zeros(2, dtype=torch.int64)
        self.buf.share_memory_()
        self.meta.share_memory_()

    @torch.no_grad()
    def publish(self, flat: torch.Tensor, iteration: int) -> None:
        seq = int(self.meta[0])
        self.meta[0] = seq + 1          # odd: write in progress
        self.buf.copy_(flat.detach().reshape(-1).cpu())
        self.meta[1] = iteration
        self.meta[0] = seq + 2          # even: stable

    @torch.no_grad()
    def iteration(self) -> int:
        return int(self.meta[1])

    @torch.no_grad()
    def read(self, out: torch.Tensor, last_iteration: int) -> Optional[int]:
        """Copy into ``out`` if a newer stable snapshot exists.

        Returns the new iteration, or None if unchanged/not yet stable.
        """
        for _ in range(8):  # bounded seqlock retry
            s0 = int(self.meta[0])
            it = int(self.meta[1])
            if s0 % 2 == 1 or it == last_iteration:
                return None
            out.reshape(-1).copy_(self.buf)
            if int(self.meta[0]) == s0:
                return it
        return None
