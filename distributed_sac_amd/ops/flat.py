"""Flat parameter/gradient storage + fused Adam.

MI355X-first design decision: every optimizer group's parameters live as
views into ONE contiguous fp32 buffer, with gradients accumulating into a
matching flat buffer.  Consequences:

- the optimizer step is ONE multi-element HIP kernel (``adam_step_``) instead
  of the reference's per-tensor torch.optim.Adam loop (K9 in SURVEY §2.6);
- the Polyak target update is ONE kernel over the flat pair (K10);
- data-parallel gradient all-reduce is ONE RCCL message per group over xGMI
  (latency-dominated at these sizes — SURVEY §2.7 MI355X mapping).

``FusedAdam`` emits/consumes ``torch.optim.Adam``-format state dicts so
reference checkpoints (learner.save_checkpoint) round-trip.
"""

from __future__ import annotations

import math
from typing import Dict, Iterable, List, Optional

import torch
import torch.nn as nn

from . import has_native, native, native_enabled


def _flatten_into(params: List[torch.Tensor]) -> torch.Tensor:
    numel = sum(p.numel() for p in params)
    if numel == 0:
        return torch.empty(0)
    flat = torch.empty(numel, dtype=torch.float32, device=params[0].device)
    off = 0
    for p in params:
        n = p.numel()
        flat[off:off + n].copy_(p.detach().reshape(-1))
        off += n
    return flat


class FlatParams:
    """Re-homes a list of nn.Parameters into one flat fp32 buffer.

    After construction each parameter's ``.data`` is a view of
    :attr:`flat_data` and its ``.grad`` a view of :attr:`flat_grad`;
    autograd accumulates straight into the flat gradient.
    """

    def __init__(self, params: Iterable[nn.Parameter], with_grad: bool = True):
        self.params: List[nn.Parameter] = [p for p in params]
        assert len(self.params) > 0, "empty parameter group"
        dev = self.params[0].device
        for p in self.params:
            assert p.device == dev, "all group params must share a device"
            assert p.dtype == torch.float32, "flat groups are fp32"
        self.flat_data = _flatten_into([p.data for p in self.params])
        self.flat_grad = (torch.zeros_like(self.flat_data) if with_grad else None)
        off = 0
        self.offsets: List[int] = []
        for p in self.params:
            n = p.numel()
            self.offsets.append(off)
            p.data = self.flat_data[off:off + n].view_as(p.data)
            if with_grad:
                p.grad = self.flat_grad[off:off + n].view_as(p.data)
            off += n
        self.numel = off

    def zero_grad(self) -> None:
        if self.flat_grad is not None:
            self.flat_grad.zero_()

    def adopt_grad_arena(self, arena: torch.Tensor, offset: int) -> int:
        """Re-home this group's flat gradient into a shared arena slice so
        several groups can all-reduce as ONE message.  Returns the next
        free arena offset."""
        n = self.numel
        new = arena[offset:offset + n]
        if self.flat_grad is not None:
            new.copy_(self.flat_grad)
        self.flat_grad = new
        self.rebind_grads()
        return offset + n

    def rebind_grads(self) -> None:
        """Re-point .grad views (autograd can replace .grad if it was None)."""
        for p, off in zip(self.params, self.offsets):
            g = self.flat_grad[off:off + p.numel()].view_as(p.data)
            if p.grad is not g:
                p.grad = g

    def view_like(self, p: nn.Parameter) -> torch.Tensor:
        i = self.params.index(p)
        off = self.offsets[i]
        return self.flat_data[off:off + p.numel()].view_as(p.data)


def flat_polyak_(target: FlatParams, source: FlatParams, tau: float,
                 mirror: Optional[torch.Tensor] = None) -> None:
    """target <- tau*source + (1-tau)*target over whole flat buffers; with
    ``mirror`` set, the target's bf16 compute mirror is refreshed in the
    same kernel (no separate cast launch).

    Requires identical parameter ordering (enforced by matching numel).
    """
    assert target.numel == source.numel
    t, s = target.flat_data, source.flat_data
    if t.is_cuda and native_enabled() and has_native():
        if mirror is not None:
            native().polyak_(t, s, float(tau), mirror)
        else:
            native().polyak_(t, s, float(tau))
        return
    t.mul_(1.0 - tau).add_(s, alpha=tau)
    if mirror is not None:
        mirror.copy_(t)


class FusedAdam:
    """Adam over a FlatParams group as one fused kernel.

    Matches torch.optim.Adam numerics exactly (bias-corrected, eps outside
    the sqrt of the bias-corrected v — torch default, weight_decay=0) and
    speaks torch.optim.Adam state_dict format for checkpoint compatibility
    with the reference (learner.save_checkpoint stores
    ``optimizer.state_dict()``).
    """

    def __init__(self, group: FlatParams, lr: float,
                 betas=(0.9, 0.999), eps: float = 1e-8, ref_params=None):
        self.group = group
        # (de)serialization order: the reference checkpoints index optimizer
        # state by the REFERENCE's parameter order (e.g. chain(critic1,
        # critic2)), which may differ from the flat group's layout (we
        # interleave layer pairs for the stacked twin-GEMM views).
        self.ref_params = (list(ref_params) if ref_params is not None
                           else list(group.params))
        self._ref_offsets = []
        for rp in self.ref_params:
            i = next(j for j, q in enumerate(group.params) if q is rp)
            self._ref_offsets.append(group.offsets[i])
        self.lr = float(lr)
        self.betas = (float(betas[0]), float(betas[1]))
        self.eps = float(eps)
        self._step_count = 0
        self.exp_avg = torch.zeros_like(group.flat_data)
        self.exp_avg_sq = torch.zeros_like(group.flat_data)
        # hipGraph-capturable path: the step counter + bias-correction
        # coefficients live on-device ({step, step_size, inv_sqrt_bc2}) so a
        # captured update keeps correct Adam bias correction across replays.
        self._dev_state = None
        # optional bf16 compute mirror refreshed in the SAME adam kernel
        # (set by the engine when mixed precision is on)
        self.bf16_mirror: Optional[torch.Tensor] = None
        if group.flat_data.is_cuda and native_enabled() and has_native():
            self._dev_state = torch.zeros(3, device=group.flat_data.device)

    @property
    def step_count(self) -> int:
        if self._dev_state is not None:
            return int(self._dev_state[0].item())
        return self._step_count

    @step_count.setter
    def step_count(self, v: int) -> None:
        self._step_count = int(v)
        if self._dev_state is not None:
            with torch.no_grad():
                self._dev_state[0] = float(v)

    def zero_grad(self) -> None:
        self.group.zero_grad()

    @torch.no_grad()
    def step(self, pre_prologed: bool = False) -> None:
        p, g = self.group.flat_data, self.group.flat_grad
        m, v = self.exp_avg, self.exp_avg_sq
        b1, b2 = self.betas
        if self._dev_state is not None:
            native().adam_step_dev_(p, g, m, v, self._dev_state,
                                    self.lr, b1, b2, self.eps,
                                    self.bf16_mirror,
                                    1 if pre_prologed else 0)
            return
        self._step_count += 1
        bc1 = 1 - b1 ** self._step_count
        bc2 = 1 - b2 ** self._step_count
        m.mul_(b1).add_(g, alpha=1 - b1)
        v.mul_(b2).addcmul_(g, g, value=1 - b2)
        denom = (v.sqrt() / math.sqrt(bc2)).add_(self.eps)
        p.addcdiv_(m, denom, value=-self.lr / bc1)

    @staticmethod
    @torch.no_grad()
    def step_many(opts: List["FusedAdam"], rng_bump=None,
                  pre_prologed: bool = False) -> None:
        """Step up to 3 optimizers in ONE fused kernel launch (same betas/
        eps; device-resident step state required).  rng_bump: optional
        int64 counter the prolog kernel increments once — the per-update
        bump for the counter-based device RNG (race-free: the prolog is a
        single-block kernel and every RNG consumer runs in other
        launches)."""
        live = [o for o in opts if o is not None]
        if (1 <= len(live) <= 3
                and all(o._dev_state is not None for o in live)
                and all(o.betas == live[0].betas and o.eps == live[0].eps
                        for o in live)):
            native().adam_step_multi_(
                [o.group.flat_data for o in live],
                [o.group.flat_grad for o in live],
                [o.exp_avg for o in live],
                [o.exp_avg_sq for o in live],
                [o._dev_state for o in live],
                [o.lr for o in live],
                live[0].betas[0], live[0].betas[1], live[0].eps,
                [o.bf16_mirror if o.bf16_mirror is not None
                 else torch.Tensor() for o in live],
                rng_bump, 1 if pre_prologed else 0)
            return
        for o in live:
            o.step()

    # -- torch.optim.Adam-compatible (de)serialization ---------------------

    def state_dict(self) -> Dict:
        state = {}
        for i, (p, off) in enumerate(zip(self.ref_params, self._ref_offsets)):
            n = p.numel()
            state[i] = {
                "step": torch.tensor(float(self.step_count)),
                "exp_avg": self.exp_avg[off:off + n].view_as(p.data).clone(),
                "exp_avg_sq": self.exp_avg_sq[off:off + n].view_as(p.data).clone(),
            }
        group = {
            "lr": self.lr, "betas": self.betas, "eps": self.eps,
            "weight_decay": 0, "amsgrad": False, "maximize": False,
            "foreach": None, "capturable": False, "differentiable": False,
            "fused": None, "params": list(range(len(self.ref_params))),
        }
        return {"state": state, "param_groups": [group]}

    def load_state_dict(self, sd: Dict) -> None:
        groups = sd["param_groups"]
        assert len(groups) == 1, "FusedAdam holds one param group"
        g0 = groups[0]
        self.lr = float(g0.get("lr", self.lr))
        if "betas" in g0:
            self.betas = tuple(float(b) for b in g0["betas"])
        self.eps = float(g0.get("eps", self.eps))
        state = sd["state"]
        if not state:
            return
        steps = []
        for i, (p, off) in enumerate(zip(self.ref_params, self._ref_offsets)):
            key = i if i in state else str(i)
            if key not in state:
                continue
            st = state[key]
            n = p.numel()
            self.exp_avg[off:off + n].copy_(
                st["exp_avg"].reshape(-1).to(self.exp_avg.device))
            self.exp_avg_sq[off:off + n].copy_(
                st["exp_avg_sq"].reshape(-1).to(self.exp_avg_sq.device))
            s = st["step"]
            steps.append(int(s.item() if torch.is_tensor(s) else s))
        if steps:
            self.step_count = max(steps)
