"""SACEngine — the learner's update step for all variants.

Implements the exact update math of the reference learners:

- 'sac' / 'vsac' (LunarLander_Distributed_SAC/src/learner.py:203-239,
  MT1_Distributed_VSAC/src/learner.py): LL-style actor, two separate
  critics + two targets, scalar log_alpha, H_bar = -action_dim.
- 'mtsac' (MT10_Distributed_MTSAC/src/learner.py:253-325): MT-style actor,
  one twin-Q critic + target, per-task log_alpha vector, per-sample alpha
  gather, optional task-weighted losses.
- 'care' lives in :mod:`.care` (adds the context/state encoder machinery).

MI355X-first mechanics under the reference math:

- every optimizer group's params/grads live in ONE flat fp32 buffer
  (:class:`~distributed_sac_amd.ops.flat.FlatParams`), so Adam is one fused
  kernel, the Polyak target update is one kernel, and data-parallel gradient
  all-reduce is one RCCL message per group;
- forward/backward run through the fused HIP MLP + squashed-Gaussian ops;
- the whole update is hipGraph-capturable (fixed shapes, graph-safe RNG) —
  see :meth:`SACEngine.maybe_capture`.
"""

from __future__ import annotations

from typing import Dict, Optional

import torch
import torch.nn as nn

from ..config import SACConfig
from ..models import Actor, Critic, LLActor, LLCritic
from ..ops import functional as Fops
from ..ops.flat import FlatParams, FusedAdam, flat_polyak_


class SACEngine:
    """Holds models/optimizers and performs one SAC gradient update."""

    def __init__(self, cfg: SACConfig, device: torch.device | str = "cpu",
                 precision: Optional[str] = None):
        import os as _os
        self.precision = (precision or
                          _os.environ.get("DSAC_PRECISION", "fp32"))
        self.cfg = cfg
        self.variant = cfg.variant
        self.device = torch.device(device)
        self.gamma = cfg.gamma
        self.tau = cfg.tau
        self.reward_scale = cfg.reward_scale
        self.use_weighted_loss = cfg.use_weighted_loss and cfg.variant == "mtsac"
        # strict-parity mode reproducing the reference's degenerate
        # (B,)x(B,1)->(B,B) weighted-loss broadcast (docs/PARITY.md);
        # supported on the torch path only — the fused HIP kernels
        # implement the corrected per-sample weighting
        self.degenerate_w = \
            getattr(cfg, "weighted_loss_mode", "corrected") == "reference"
        self.num_tasks = cfg.num_tasks if cfg.variant in ("mtsac", "care") else 1
        self.update_iteration = 0
        self.total_step = 0
        self._graph = None
        self._eps_queue: Optional[list] = None  # test hook: deterministic eps
        self.ddp = None  # optional DataParallelGroup (set via attach_ddp)
        self._build_models()
        self._build_optimizers()

    def _next_eps(self, like: torch.Tensor) -> torch.Tensor:
        if self._eps_queue:
            return self._eps_queue.pop(0).to(like.device)
        return torch.randn_like(like)

    def _sample(self, states: torch.Tensor):
        """Actor sampling with squashed-Gaussian op (both actor families)."""
        mu, log_std_raw = self.actor(states)
        eps = self._next_eps(mu)
        return Fops.squashed_gaussian(mu, log_std_raw, eps, self.actor.k)

    # ------------------------------------------------------------------
    def _build_models(self) -> None:
        cfg, dev = self.cfg, self.device
        if self.variant in ("sac", "vsac"):
            self.actor = LLActor(cfg.state_dim, cfg.action_dim,
                                 cfg.actor_hidden_dim, cfg.action_bound).to(dev)
            self.local_critic_1 = LLCritic(cfg.state_dim, cfg.action_dim,
                                           cfg.critic_hidden_dim).to(dev)
            self.local_critic_2 = LLCritic(cfg.state_dim, cfg.action_dim,
                                           cfg.critic_hidden_dim).to(dev)
            self.target_critic_1 = LLCritic(cfg.state_dim, cfg.action_dim,
                                            cfg.critic_hidden_dim).to(dev)
            self.target_critic_2 = LLCritic(cfg.state_dim, cfg.action_dim,
                                            cfg.critic_hidden_dim).to(dev)
            self.log_alpha = nn.Parameter(torch.tensor(
                [float(cfg.log_alpha)], device=dev))
        elif self.variant == "mtsac":
            self.actor = Actor(cfg.state_dim, cfg.action_dim,
                               cfg.actor_hidden_dim, cfg.action_bound,
                               num_tasks=cfg.num_tasks).to(dev)
            self.local_critic = Critic(cfg.state_dim, cfg.action_dim,
                                       cfg.critic_hidden_dim,
                                       num_tasks=cfg.num_tasks).to(dev)
            self.target_critic = Critic(cfg.state_dim, cfg.action_dim,
                                        cfg.critic_hidden_dim,
                                        num_tasks=cfg.num_tasks).to(dev)
            # per-task log_alpha (reference MT10…MTSAC/src/learner.py:115-121)
            self.log_alpha = nn.Parameter(torch.full(
                (cfg.num_tasks,), float(cfg.log_alpha), device=dev))
        else:
            raise ValueError(f"variant {self.variant} not handled here")
        self.H_bar = torch.tensor([-float(cfg.action_dim)], device=dev)
        self.H_bar_f = -float(cfg.action_dim)  # python scalar: graph-safe
        self.alpha = self.log_alpha.exp().detach()

    @staticmethod
    def _critic_linears(c1_or_critic, c2=None):
        """Per-layer Linear pairs (Q1_i, Q2_i) across both critic styles."""
        if c2 is not None:  # LL style: two modules
            l1 = [c1_or_critic.first_layer] + list(c1_or_critic.layer_module)
            l2 = [c2.first_layer] + list(c2.layer_module)
        else:  # MT style: one module with Q_function_1/2 Sequentials
            import torch.nn as nn_
            l1 = [m for m in c1_or_critic.Q_function_1 if isinstance(m, nn_.Linear)]
            l2 = [m for m in c1_or_critic.Q_function_2 if isinstance(m, nn_.Linear)]
        return list(zip(l1, l2))

    def _critic_layer_pairs(self, target: bool = False):
        if self.variant in ("sac", "vsac"):
            a = self.target_critic_1 if target else self.local_critic_1
            b = self.target_critic_2 if target else self.local_critic_2
            return self._critic_linears(a, b)
        return self._critic_linears(self.target_critic if target
                                    else self.local_critic)

    def _critic_params(self):
        """Interleaved per-layer order [Q1.w_i, Q2.w_i, Q1.b_i, Q2.b_i] so
        the flat buffer contains contiguous stacked [2,N,K] weights for the
        grouped twin-GEMM kernels."""
        out = []
        for l1, l2 in self._critic_layer_pairs():
            out += [l1.weight, l2.weight, l1.bias, l2.bias]
        return out

    def _target_params(self):
        out = []
        for l1, l2 in self._critic_layer_pairs(target=True):
            out += [l1.weight, l2.weight, l1.bias, l2.bias]
        return out

    def _build_twin_stacks(self, group, pairs, with_grad: bool):
        """Stacked [2,N,K]/[2,N] leaf views into a flat buffer, with .grad
        views into the flat gradient so autograd accumulates straight into
        the all-reduce/Adam buffer."""
        ws, bs = [], []
        for l1, l2 in pairs:
            for (pa, pb, shape, dest) in (
                    (l1.weight, l2.weight, l1.weight.shape, ws),
                    (l1.bias, l2.bias, l1.bias.shape, bs)):
                ia = next(i for i, q in enumerate(group.params) if q is pa)
                ib = next(i for i, q in enumerate(group.params) if q is pb)
                off = group.offsets[ia]
                n = pa.numel()
                assert group.offsets[ib] == off + n, "stacking needs adjacency"
                view = group.flat_data[off:off + 2 * n].view(2, *shape)
                stack = view.detach()
                if with_grad:
                    stack.requires_grad_(True)
                    stack.grad = group.flat_grad[off:off + 2 * n].view(2, *shape)
                dest.append(stack)
        return ws, bs

    def _build_optimizers(self) -> None:
        cfg = self.cfg
        self.actor_group = FlatParams(self.actor.parameters())
        self.critic_group = FlatParams(self._critic_params())
        self.target_group = FlatParams(self._target_params(), with_grad=False)
        self.alpha_group = FlatParams([self.log_alpha])
        self.actor_optimizer = FusedAdam(self.actor_group, lr=cfg.lr_actor)
        # checkpoint state is indexed in the REFERENCE optimizer order:
        # chain(critic1, critic2) for LL/VSAC (learner.build_optimizer),
        # local_critic.parameters() for MT — not our interleaved layout
        if self.variant in ("sac", "vsac"):
            ref_order = (list(self.local_critic_1.parameters())
                         + list(self.local_critic_2.parameters()))
        else:
            ref_order = list(self.local_critic.parameters())
        self.critic_optimizer = FusedAdam(self.critic_group,
                                          lr=cfg.lr_critic,
                                          ref_params=ref_order)
        # reference uses lr_actor for log_alpha (learner.py build_optimizer)
        self.log_alpha_optimizer = FusedAdam(self.alpha_group, lr=cfg.lr_actor)
        self._twin_local = self._build_twin_stacks(
            self.critic_group, self._critic_layer_pairs(), with_grad=True)
        self._twin_target = self._build_twin_stacks(
            self.target_group, self._critic_layer_pairs(target=True),
            with_grad=False)
        # frozen aliases of the local critic stacks (same storage, no
        # requires_grad): the actor pass needs dQ/d(action) but the critic
        # weight grads it would deposit are discarded (reference zeroes them
        # next update) — skipping their computation saves 4 dW GEMMs/step.
        self._twin_local_frozen = ([w.detach() for w in self._twin_local[0]],
                                   [b.detach() for b in self._twin_local[1]])
        # actor+alpha share one gradient arena: their all-reduce (both due
        # between the same backward and the fused Adam) is ONE RCCL message
        self._aa_arena = torch.zeros(
            self.actor_group.numel + self.alpha_group.numel,
            device=self.device)
        off = self.actor_group.adopt_grad_arena(self._aa_arena, 0)
        self.alpha_group.adopt_grad_arena(self._aa_arena, off)
        self._init_bf16_mirrors()
        self.hard_copy_targets()

    # -- bf16 mixed precision: fp32 masters + bf16 compute mirrors --------
    def _init_bf16_mirrors(self) -> None:
        self._bf16 = (self.precision == "bf16"
                      and self.device.type == "cuda")
        if not self._bf16:
            return
        dev = self.device
        # persistent zero-initialized workspace for the one-launch loss
        # reductions (per-block partial slots + self-resetting ticket)
        self._loss_ws = torch.zeros(256, device=dev)
        self._aloss_ws = torch.zeros(256, device=dev)
        # persistent counter for the in-kernel RNG, seeded from the torch
        # seed so runs stay reproducible end-to-end
        self._rng_ctr = torch.tensor(
            [int(torch.initial_seed()) & 0x7FFFFFFF],
            dtype=torch.int64, device=dev)
        self._critic_bf16 = torch.empty(self.critic_group.numel,
                                        dtype=torch.bfloat16, device=dev)
        self._target_bf16 = torch.empty(self.target_group.numel,
                                        dtype=torch.bfloat16, device=dev)
        self._actor_bf16 = torch.empty(self.actor_group.numel,
                                       dtype=torch.bfloat16, device=dev)
        self._twin_local_bf16 = self._stack_views(self.critic_group,
                                                  self._critic_bf16,
                                                  self._critic_layer_pairs())
        self._twin_target_bf16 = self._stack_views(
            self.target_group, self._target_bf16,
            self._critic_layer_pairs(target=True))
        self.critic_optimizer.bf16_mirror = self._critic_bf16
        self.actor_optimizer.bf16_mirror = self._actor_bf16
        ws, _ = self._actor_weights()
        self._actor_ws_bf16 = []
        for w in ws:
            i = next(j for j, q in enumerate(self.actor_group.params)
                     if q is w)
            off = self.actor_group.offsets[i]
            self._actor_ws_bf16.append(
                self._actor_bf16[off:off + w.numel()].view_as(w))
        # TRANSPOSED weight mirrors (shape carriers for the dx-chain
        # binding; the VALUES now come from the fragment-packed mirrors
        # below, so these are never refreshed on the packed path)
        self._twin_local_wt = [
            torch.empty(w.shape[0], w.shape[2], w.shape[1],
                        dtype=torch.bfloat16, device=dev)
            for w in self._twin_local_bf16]
        self._actor_wt = [
            torch.empty(w.shape[1], w.shape[0], dtype=torch.bfloat16,
                        device=dev) for w in self._actor_ws_bf16]

        # FRAGMENT-PACKED weight mirrors (k_bf16_pack_frag): the chain
        # kernels' B-operand loads become base + lane*16 unit-stride
        # (one contiguous 1 KiB line per wave) instead of 16 scattered
        # cache lines — re-packed per update (pre-step set at seg1,
        # post-critic-Adam set in seg2)
        def _fp(w, G):
            K = w.shape[-1]
            N = w.numel() // (G * K)
            return torch.empty(
                G * ((N + 15) // 16) * ((K + 31) // 32) * 512,
                dtype=torch.bfloat16, device=dev)

        def _dxp(w, G):
            K = w.shape[-1]
            N = w.numel() // (G * K)
            return torch.empty(
                G * ((K + 15) // 16) * ((N + 31) // 32) * 512,
                dtype=torch.bfloat16, device=dev)

        self._twin_local_fp = [_fp(w, 2) for w in self._twin_local_bf16]
        self._twin_target_fp = [_fp(w, 2) for w in self._twin_target_bf16]
        self._actor_fp = [_fp(w, 1) for w in self._actor_ws_bf16]
        self._twin_local_dxp = [_dxp(w, 2) for w in self._twin_local_bf16]
        self._actor_dxp = [_dxp(w, 1) for w in self._actor_ws_bf16]
        self.refresh_bf16()

    @staticmethod
    def _stack_views(group, mirror, pairs):
        ws = []
        for l1, l2 in pairs:
            ia = next(i for i, q in enumerate(group.params)
                      if q is l1.weight)
            off = group.offsets[ia]
            n = l1.weight.numel()
            ws.append(mirror[off:off + 2 * n].view(2, *l1.weight.shape))
        return ws

    @torch.no_grad()
    def refresh_bf16(self, which: str = "all") -> None:
        if not getattr(self, "_bf16", False):
            return
        if which in ("all", "critic"):
            self._critic_bf16.copy_(self.critic_group.flat_data)
        if which in ("all", "target"):
            self._target_bf16.copy_(self.target_group.flat_data)
        if which in ("all", "actor"):
            self._actor_bf16.copy_(self.actor_group.flat_data)

    def attach_ddp(self, ddp) -> None:
        """Join a data-parallel group: sync replicas, then every update
        all-reduces the flat gradient buffers (one RCCL message each)."""
        self.ddp = ddp
        if ddp is not None and ddp.enabled:
            ddp.broadcast_params(self.actor_group.flat_data)
            ddp.broadcast_params(self.critic_group.flat_data)
            ddp.broadcast_params(self.alpha_group.flat_data)
            self.hard_copy_targets()
            self.refresh_bf16()

    @torch.no_grad()
    def publish_params(self) -> torch.Tensor:
        """Flat weight vector for the rollout-side snapshot (matches
        workers.player.policy_params order)."""
        return self.actor_group.flat_data

    @torch.no_grad()
    def hard_copy_targets(self) -> None:
        """targets <- critics (reference soft_update tau=1.0 at run start)."""
        flat_polyak_(self.target_group, self.critic_group, 1.0)
        self.refresh_bf16("target")

    def zero_grad(self) -> None:
        self.actor_group.zero_grad()
        self.critic_group.zero_grad()
        self.alpha_group.zero_grad()
        # autograd may have detached .grad views (e.g. after state_dict load)
        self.actor_group.rebind_grads()
        self.critic_group.rebind_grads()
        self.alpha_group.rebind_grads()

    # ------------------------------------------------------------------
    def _use_fused(self, t: torch.Tensor) -> bool:
        from ..ops import has_native, native_enabled
        return t.is_cuda and native_enabled() and has_native()

    def _critic_q(self, states, actions):
        if self._use_fused(states):
            x = torch.cat([states, actions], dim=-1)
            if getattr(self, "_bf16", False):
                return Fops.twin_mlp_forward_bf16(x, *self._twin_local,
                                                  self._twin_local_bf16)
            return Fops.twin_mlp_forward(x, *self._twin_local)
        if self.variant in ("sac", "vsac"):
            return (self.local_critic_1(states, actions),
                    self.local_critic_2(states, actions))
        return self.local_critic(states, actions)

    def _target_q(self, states, actions):
        if self._use_fused(states):
            x = torch.cat([states, actions], dim=-1)
            if getattr(self, "_bf16", False):
                return Fops.twin_mlp_forward_bf16(x, *self._twin_target,
                                                  self._twin_target_bf16)
            return Fops.twin_mlp_forward(x, *self._twin_target)
        if self.variant in ("sac", "vsac"):
            return (self.target_critic_1(states, actions),
                    self.target_critic_2(states, actions))
        return self.target_critic(states, actions)

    def _per_sample_alpha(self, mtobss: torch.Tensor) -> torch.Tensor:
        """mtsac: per-sample alpha via one-hot gather; sac/vsac: scalar."""
        if self.variant == "mtsac":
            one_hots = mtobss[:, -self.num_tasks:]
            return Fops.gather_log_alpha(one_hots, self.log_alpha).exp().detach()
        return self.log_alpha.exp().detach()

    def update(self, batch: Dict[str, torch.Tensor]) -> Dict[str, float]:
        """One SAC gradient update; returns scalar metrics (syncs)."""
        out = self.update_tensors(batch)
        self.update_iteration += 1
        return {k: float(v.detach()) for k, v in out.items()}

    def update_tensors(self, batch: Dict[str, torch.Tensor]) -> Dict[str, torch.Tensor]:
        """One SAC gradient update, returning tensor metrics (no host sync
        — hipGraph-capturable).

        Order (identical to reference update/update_SAC): TD target →
        critic step → actor step → alpha step → Polyak.
        """
        if self._use_fused(batch["states"]):
            if self.degenerate_w and self.use_weighted_loss:
                raise RuntimeError(
                    "weighted_loss_mode='reference' (degenerate broadcast) "
                    "is a CPU strict-parity mode; the fused HIP kernels "
                    "implement corrected weighting (docs/PARITY.md)")
            return self._update_tensors_fused(batch)
        states = batch["states"]
        actions = batch["actions"]
        rewards = batch["rewards"]
        next_states = batch["next_states"]
        dones = batch["dones"]

        alpha = self._per_sample_alpha(states)
        self.zero_grad()

        # --- TD target (no grad) -------------------------------------
        with torch.no_grad():
            next_actions, next_log_probs, _ = self._sample(next_states)
            q1_t, q2_t = self._target_q(next_states, next_actions)
            y = Fops.td_target(rewards, dones, q1_t, q2_t, next_log_probs,
                               alpha, self.gamma, self.reward_scale)

        # --- critic step ---------------------------------------------
        if self.variant in ("sac", "vsac"):
            q_loss = (self.local_critic_1.cal_loss(states, actions, y)
                      + self.local_critic_2.cal_loss(states, actions, y))
        else:
            l1, l2 = self.local_critic.cal_loss(
                states, actions, y,
                use_weighted_loss=self.use_weighted_loss,
                alphas=self.log_alpha.exp().detach(),
                degenerate=self.degenerate_w)
            q_loss = l1 + l2
        q_loss.backward()
        if self.ddp is not None:
            self.ddp.allreduce_grad_(self.critic_group.flat_grad)
        self.critic_optimizer.step()

        # --- actor step ----------------------------------------------
        sampled_actions, log_probs, log_stds = self._sample(states)
        q1, q2 = self._critic_q(states, sampled_actions)
        q_min = torch.min(q1, q2)
        if self.variant == "mtsac":
            policy_loss = self.actor.cal_loss(
                log_probs, q_min, alpha,
                use_weighted_loss=self.use_weighted_loss, mtobss=states,
                alphas=self.log_alpha.exp().detach(),
                degenerate=self.degenerate_w)
        else:
            policy_loss = self.actor.cal_loss(log_probs, q_min, alpha)
        policy_loss.backward()
        if self.ddp is not None:
            self.ddp.allreduce_grad_(self.actor_group.flat_grad)
        self.actor_optimizer.step()

        # --- entropy diagnostic (reference learner.py:311-312) --------
        entropy = Fops.entropy_from_log_std(log_stds)

        # --- temperature step -----------------------------------------
        if self.variant == "mtsac":
            log_alpha_g = Fops.gather_log_alpha(
                states[:, -self.num_tasks:], self.log_alpha)
            loss_log_alpha = -(log_alpha_g * (log_probs.detach() + self.H_bar)).mean()
        else:
            loss_log_alpha = -(self.log_alpha * (log_probs.detach() + self.H_bar)).mean()
        loss_log_alpha.backward()
        if self.ddp is not None:
            self.ddp.allreduce_grad_(self.alpha_group.flat_grad)
        self.log_alpha_optimizer.step()
        self.alpha = self.log_alpha.exp().detach()

        # --- Polyak target update -------------------------------------
        flat_polyak_(self.target_group, self.critic_group, self.tau)

        return {
            "critic_loss": q_loss.detach(),
            "actor_loss": policy_loss.detach(),
            "alpha_loss": loss_log_alpha.detach(),
            "entropy": entropy.detach(),
        }

    def _actor_weights(self):
        if self.variant in ("sac", "vsac"):
            ws = [m.weight for m in self.actor.layer_intermediate] \
                + [self.actor.mu_log_std_layer.weight]
            bs = [m.bias for m in self.actor.layer_intermediate] \
                + [self.actor.mu_log_std_layer.bias]
            return ws, bs
        import torch.nn as nn_
        lins = [m for m in self.actor.mu_log_std_layer
                if isinstance(m, nn_.Linear)]
        return [m.weight for m in lins], [m.bias for m in lins]

    @torch.no_grad()
    def _mlp_fwd_manual(self, x_f32, ws_bf16, bs_f32):
        from ..ops import native
        ext = native()
        xh = x_f32.to(torch.bfloat16)
        acts = [xh]
        h = xh
        n = len(ws_bf16)
        for i in range(n):
            last = i == n - 1
            h = ext.linear_act_fwd_bf16(h, ws_bf16[i], bs_f32[i].contiguous(),
                                        0 if last else 1, 1,
                                        1 if last else 0)
            acts.append(h)
        return h, acts

    @torch.no_grad()
    def _twin_fwd_manual(self, x_f32, ws_bf16, bs_f32):
        from ..ops import native
        ext = native()
        xh = x_f32.to(torch.bfloat16)
        acts = [xh]
        h = xh
        n = len(ws_bf16)
        for i in range(n):
            last = i == n - 1
            h = ext.linear_act_fwd_bf16(h, ws_bf16[i], bs_f32[i].contiguous(),
                                        0 if last else 1, 2,
                                        1 if last else 0)
            acts.append(h)
        return h[0], h[1], acts

    def _dw_arena(self, name: str, numel: int, B: int):
        """[S, group_numel] fp32 split-K arena for one backward phase:
        every layer's partials land at its flat-gradient offsets, and ONE
        k_reduce_arena launch folds the whole phase's gradient (instead of
        one reduce per layer).  chunk targets ~8 batch splits, rounded to
        the 64-row GEMM tile."""
        store = getattr(self, "_dw_arenas", None)
        if store is None:
            store = self._dw_arenas = {}
        key = (name, numel, B)
        if key not in store:
            if name.endswith("@rowblocks"):
                # one arena row per 64-row block (fused narrow backward)
                S = (B + 63) // 64
                chunk = 64
            else:
                chunk = ((B + 7) // 8 + 63) // 64 * 64
                S = (B + chunk - 1) // chunk
            store[key] = (torch.empty(S, numel, device=self.device),
                          S, chunk)
        return store[key]

    @staticmethod
    def _mlp_bwd_arena(ext, dy, acts, wsh, ws_grad, bs_grad, flat_grad,
                       arena, S, chunk, G=1, transpose_w=0):
        """Backward an act-fwd chain; dW/db partials go to the phase arena
        at the layer's flat-grad offsets (fold happens once per phase)."""
        n = len(wsh)
        base = flat_grad.data_ptr()
        for i in range(n - 1, -1, -1):
            act = 1 if i < n - 1 else 0
            yout = acts[i + 1] if i < n - 1 else acts[i]
            ext.linear_bwd_dwdb_arena(
                dy, acts[i], yout, act, G, arena,
                (ws_grad[i].data_ptr() - base) // 4,
                (bs_grad[i].data_ptr() - base) // 4, S, chunk, transpose_w)
            if i > 0:
                dy = ext.linear_bwd_dx_bf16(dy, wsh[i], yout, act, G, 0)
        return dy

    @torch.no_grad()
    def _update_tensors_manual(self, batch):
        """bf16 path with HAND-ROLLED backward: every gradient-producing
        kernel writes straight into the flat fp32 gradient buffers (no
        autograd bookkeeping, no AccumulateGrad adds, no zero_grad except
        the tiny atomically-accumulated alpha grad).  Numerically identical
        to the autograd bf16 path (same kernels, same order) — verified by
        the manual-vs-autograd GPU test.

        Structured as three SEGMENTS split at the two DP all-reduce
        boundaries so the data-parallel path can hipGraph-capture each
        segment and run the (non-capturable) RCCL collectives eagerly
        between replays — see :meth:`capture_dp`:
          seg1: forwards + TD target + critic backward  -> critic grad
          seg2: critic Adam + actor/alpha forwards+backwards -> aa grads
          seg3: actor+alpha fused Adam + Polyak
        """
        self._manual_seg1(batch)
        if self.ddp is not None:
            self.ddp.allreduce_grad_(self.critic_group.flat_grad)
        self._manual_seg2()
        if self.ddp is not None:
            self.ddp.allreduce_grad_(self._aa_arena)
        return self._manual_seg3()

    @property
    def _use_chain(self) -> bool:
        """Fused MLP-chain forward kernels (one launch per chain,
        in-kernel f32 concat+cast input — kills the per-layer launch
        latency plus the cat/cast launches; round-2).  DSAC_CHAIN=0
        restores the per-layer path."""
        import os as _os
        return _os.environ.get("DSAC_CHAIN", "1") == "1"

    @property
    def _use_krng(self) -> bool:
        """Counter-based device RNG inside the squash / replay-sample
        kernels (round 2): replaces the torch rand+randn launches and the
        hipGraph RNG-offset bookkeeping kernels.  DSAC_KRNG=0 restores
        torch's philox draws."""
        import os as _os
        return (getattr(self, "_bf16", False)
                and getattr(self, "_rng_ctr", None) is not None
                and _os.environ.get("DSAC_KRNG", "1") == "1")

    def _adam_prolog_all(self) -> bool:
        """ONE launch advances every optimizer's Adam state (+ RNG bump)
        at the top of seg2; the individual steppers then skip their own
        prolog launches.  Returns False (fall back to per-step prologs)
        if any optimizer lacks device state or betas differ."""
        from ..ops import native
        opts = [self.critic_optimizer, self.actor_optimizer,
                self.log_alpha_optimizer]
        ctx = getattr(self, "context_encoder_optimizer", None)
        if ctx is not None:
            opts.append(ctx)
        if any(o._dev_state is None for o in opts):
            return False
        b1, b2 = opts[0].betas
        if any(o.betas != (b1, b2) or o.eps != opts[0].eps for o in opts):
            return False
        native().adam_prolog_many(
            [o._dev_state for o in opts], [o.lr for o in opts], b1, b2,
            self._rng_ctr if self._use_krng else None)
        return True

    @torch.no_grad()
    def _chain_fwd(self, x1, x2, ws_bf16, bs_f32, G, act_last=0,
                   out_f32=True, rowcat=False, save_acts=True, wps=None):
        from ..ops import native
        out = native().mlp_chain_fwd_bf16(
            x1, x2 if x2 is not None else x1.new_empty(0),
            list(ws_bf16), [b.contiguous() for b in bs_f32],
            int(act_last), int(G), 1 if out_f32 else 0, 0,
            1 if rowcat else 0, 1 if save_acts else 0,
            list(wps) if wps is not None else [])
        return out[0], [out[1]] + list(out[2:])

    @torch.no_grad()
    def _manual_seg1(self, batch):
        """Segment 1: batched actor forward + squash, TD target, critic
        loss forward + backward.  Ends with the critic flat gradient ready
        for all-reduce."""
        from ..ops import native
        ext = native()
        states = batch["states"]
        actions = batch["actions"]
        rewards = batch["rewards"]
        next_states = batch["next_states"]
        dones = batch["dones"]
        T = self.num_tasks
        use_w = self.use_weighted_loss
        B = states.shape[0]
        A = self.cfg.action_dim
        la_det = self.log_alpha.detach()
        chain = self._use_chain
        if chain:
            # ONE launch re-packs every pre-step weight set into the
            # fragment layout (critic fwd+dx, target fwd, actor fwd+dx)
            nl_c0 = len(self._twin_local_bf16)
            nl_a0 = len(self._actor_ws_bf16)
            ext.pack_weights_frag(
                list(self._twin_local_bf16) * 2
                + list(self._twin_target_bf16)
                + list(self._actor_ws_bf16) * 2,
                list(self._twin_local_fp) + list(self._twin_local_dxp)
                + list(self._twin_target_fp)
                + list(self._actor_fp) + list(self._actor_dxp),
                [2] * (3 * nl_c0) + [1] * (2 * nl_a0),
                [0] * nl_c0 + [1] * nl_c0 + [0] * nl_c0
                + [0] * nl_a0 + [1] * nl_a0)

        # ---- batched actor forward + squash --------------------------
        ws_f32, bs_f32 = self._actor_weights()
        if chain:
            out, acts_a = self._chain_fwd(next_states, states,
                                          self._actor_ws_bf16, bs_f32,
                                          G=1, rowcat=True,
                                          wps=self._actor_fp)
        else:
            x_cat = torch.cat([next_states, states], dim=0)
            out, acts_a = self._mlp_fwd_manual(x_cat, self._actor_ws_bf16,
                                               bs_f32)
        mu, lsr = out[:, :A], out[:, A:]
        krng = self._use_krng and not self._eps_queue
        if self._eps_queue:
            eps = torch.cat([self._next_eps(mu[:B]), self._next_eps(mu[:B])])
        elif krng:
            # eps GENERATED inside the squash kernel (counter RNG) and
            # written here for the backward — no randn launch
            eps = torch.empty(mu.shape, device=mu.device, dtype=mu.dtype)
        else:
            eps = torch.randn_like(mu)
        a_cat, lp_cat, tanh_u, ls_cat = ext.squashed_gaussian_fwd(
            mu, lsr, eps, float(self.actor.k),
            self._rng_ctr if krng else None)
        na, nlp = a_cat[:B], lp_cat[:B]

        # ---- TD target ------------------------------------------------
        if chain:
            yt, _ = self._chain_fwd(next_states, na,
                                    self._twin_target_bf16,
                                    self._twin_target[1], G=2,
                                    save_acts=False,
                                    wps=self._twin_target_fp)
            q1_t, q2_t = yt[0], yt[1]
        else:
            xt = torch.cat([next_states, na], dim=-1)
            q1_t, q2_t, _ = self._twin_fwd_manual(
                xt, self._twin_target_bf16, self._twin_target[1])
        y = ext.td_target_mt(rewards, dones, q1_t, q2_t, nlp, states,
                             la_det, T, self.gamma, self.reward_scale)

        # ---- critic loss + manual backward ---------------------------
        if chain:
            yq, acts_c = self._chain_fwd(states, actions,
                                         self._twin_local_bf16,
                                         self._twin_local[1], G=2,
                                         wps=self._twin_local_fp)
            q1, q2 = yq[0], yq[1]
        else:
            x = torch.cat([states, actions], dim=-1)
            q1, q2, acts_c = self._twin_fwd_manual(
                x, self._twin_local_bf16, self._twin_local[1])
        closs = ext.critic_loss_fwd(q1, q2, y, states, la_det, T,
                                    int(use_w), self._loss_ws)[0]
        dy = ext.critic_loss_bwd2(q1, q2, y, states, la_det, closs, T,
                                  int(use_w))
        wsg = [w.grad for w in self._twin_local[0]]
        bsg = [b.grad for b in self._twin_local[1]]
        fg_c = self.critic_group.flat_grad
        arena_c, S_c, ch_c = self._dw_arena("critic",
                                            self.critic_group.numel, B)
        if chain:
            nl_c = len(self._twin_local_bf16)
            empty_h = states.new_empty(0, dtype=torch.bfloat16)
            youts = [acts_c[i + 1] for i in range(nl_c - 1)] + [empty_h]
            aflags = [1] * (nl_c - 1) + [0]
            dys = ext.mlp_chain_dx_bf16(dy, list(self._twin_local_wt),
                                        youts, acts_c[0].shape[-1],
                                        aflags, 2, 1, -1,
                                        list(self._twin_local_dxp))
            base_c = fg_c.data_ptr()
            ext.dwdb_grouped_arena(
                list(dys), [acts_c[i] for i in range(nl_c)], arena_c,
                [(w.data_ptr() - base_c) // 4 for w in wsg],
                [(b.data_ptr() - base_c) // 4 for b in bsg],
                2, S_c, ch_c)
        else:
            self._mlp_bwd_arena(ext, dy, acts_c, self._twin_local_bf16,
                                wsg, bsg, fg_c, arena_c, S_c, ch_c, G=2)
        ext.reduce_arena(arena_c, fg_c, S_c)
        self._dp_st = {
            "states": states, "sa": a_cat[B:], "lp": lp_cat[B:],
            "lsr": lsr, "ls_cat": ls_cat, "eps": eps, "tanh_u": tanh_u,
            "acts_a": acts_a, "ws_f32": ws_f32, "bs_f32": bs_f32,
            "la_det": la_det, "closs": closs, "B": B,
        }

    @torch.no_grad()
    def _manual_seg2(self):
        """Segment 2: critic Adam (mirror refreshed in-kernel), actor-side
        twin forward on the post-step critic, actor/alpha losses and full
        manual backward.  Ends with the fused actor+alpha gradient arena
        ready for all-reduce."""
        from ..ops import native
        ext = native()
        st = self._dp_st
        st["prolog"] = self._adam_prolog_all()
        # adam kernel also refreshes the bf16 mirror
        self.critic_optimizer.step(pre_prologed=st["prolog"])

        states, sa, lp = st["states"], st["sa"], st["lp"]
        B = st["B"]
        T = self.num_tasks
        use_w = self.use_weighted_loss
        la_det = st["la_det"]
        ls = st["ls_cat"][B:]
        nl_a = len(self._actor_ws_bf16)
        nl_c = len(self._twin_local_bf16)
        acts_a = st["acts_a"]

        if self._use_chain:
            # re-pack the POST-Adam critic (fwd + dx) in one launch
            ext.pack_weights_frag(
                list(self._twin_local_bf16) * 2,
                list(self._twin_local_fp) + list(self._twin_local_dxp),
                [2] * (2 * nl_c), [0] * nl_c + [1] * nl_c)
            ya, acts_f = self._chain_fwd(states, sa, self._twin_local_bf16,
                                         self._twin_local[1], G=2,
                                         wps=self._twin_local_fp)
            aq1, aq2 = ya[0], ya[1]
        else:
            xa = torch.cat([states, sa], dim=-1)
            aq1, aq2, acts_f = self._twin_fwd_manual(
                xa, self._twin_local_bf16, self._twin_local[1])
        # the fwd kernel zeroes the alpha flat grad in passing (the bwd's
        # atomics target) — no separate fill launch
        al = ext.actor_alpha_loss_fwd(aq1, aq2, lp, ls, states,
                                      la_det, T, int(use_w), self.H_bar_f,
                                      self.alpha_group.flat_grad,
                                      self._aloss_ws)
        daq, dlp = ext.actor_alpha_loss_bwd2(
            aq1, aq2, lp, states, la_det, al, self.alpha_group.flat_grad,
            T, int(use_w), self.H_bar_f)
        if self._use_chain:
            empty_h = states.new_empty(0, dtype=torch.bfloat16)
            youts_f = [acts_f[i + 1] for i in range(nl_c - 1)] + [empty_h]
            outs = ext.mlp_chain_dx_bf16(
                daq, list(self._twin_local_wt), youts_f,
                acts_f[0].shape[-1], [1] * (nl_c - 1) + [0], 2, 0,
                states.shape[1], list(self._twin_local_dxp))
            dx0 = outs[-1]          # [2, B, A] fp32 (action columns only)
            dsa = dx0               # twin heads summed inside squash bwd2
        else:
            dy = daq
            for i in range(nl_c - 1, 0, -1):
                act = 1 if i < nl_c - 1 else 0
                yout = acts_f[i + 1] if i < nl_c - 1 else acts_f[i]
                dy = ext.linear_bwd_dx_bf16(dy, self._twin_local_bf16[i],
                                            yout, act, 2, 0)
            dxa = ext.linear_bwd_dx_bf16(
                dy, self._twin_local_bf16[0],
                acts_f[1] if nl_c > 1 else acts_f[0],
                1 if nl_c > 1 else 0, 2, 1)
            dsa = dxa[:, states.shape[1]:].float()
        dhead = ext.squashed_gaussian_bwd2(
            dsa, dlp, st["lsr"][B:], st["ls_cat"][B:], st["eps"][B:],
            st["tanh_u"][B:], float(self.actor.k))
        wag = [w.grad for w in st["ws_f32"]]
        bag = [b.grad for b in st["bs_f32"]]
        fg_a = self.actor_group.flat_grad
        arena_a, S_a, ch_a = self._dw_arena("actor",
                                            self.actor_group.numel, B)
        base_a = fg_a.data_ptr()
        if self._use_chain:
            empty_h = states.new_empty(0, dtype=torch.bfloat16)
            youts_a = [acts_a[i + 1][B:] for i in range(nl_a - 1)] \
                + [empty_h]
            dys_a = ext.mlp_chain_dx_bf16(
                dhead, list(self._actor_wt), youts_a,
                acts_a[0].shape[-1], [1] * (nl_a - 1) + [0], 1, 1, -1,
                list(self._actor_dxp))
            ext.dwdb_grouped_arena(
                list(dys_a), [acts_a[i][B:] for i in range(nl_a)],
                arena_a,
                [(w.data_ptr() - base_a) // 4 for w in wag],
                [(b.data_ptr() - base_a) // 4 for b in bag],
                1, S_a, ch_a)
        else:
            dy = dhead
            for i in range(nl_a - 1, -1, -1):
                act = 1 if i < nl_a - 1 else 0
                yout = (acts_a[i + 1][B:] if i < nl_a - 1
                        else acts_a[i][B:])
                ext.linear_bwd_dwdb_arena(
                    dy, acts_a[i][B:], yout, act, 1, arena_a,
                    (wag[i].data_ptr() - base_a) // 4,
                    (bag[i].data_ptr() - base_a) // 4, S_a, ch_a, 0)
                if i > 0:
                    dy = ext.linear_bwd_dx_bf16(dy, self._actor_ws_bf16[i],
                                                yout, act, 1, 1)
        ext.reduce_arena(arena_a, fg_a, S_a)
        st["al"] = al

    @torch.no_grad()
    def _manual_seg3(self):
        """Segment 3: fused actor+alpha Adam, Polyak target update."""
        st = self._dp_st
        FusedAdam.step_many([self.actor_optimizer,
                             self.log_alpha_optimizer],
                            rng_bump=(None if st.get("prolog")
                                      else (self._rng_ctr if self._use_krng
                                            else None)),
                            pre_prologed=bool(st.get("prolog")))
        # NOTE: self.alpha is refreshed lazily (checkpoint/_per_sample_alpha
        # recompute from log_alpha) — an exp() here would replay as a
        # ~5 µs kernel every captured step just for bookkeeping
        flat_polyak_(self.target_group, self.critic_group, self.tau,
                     mirror=getattr(self, "_target_bf16", None))
        closs, al = st["closs"], st["al"]
        return {
            "critic_loss": closs[6],  # summed in-kernel
            "actor_loss": al[0],
            "alpha_loss": al[2],
            "entropy": al[3],
        }

    def _update_tensors_fused(self, batch):
        """GPU path: same math/order as update_tensors, restructured for
        the hardware (numerically identical):

        - ONE actor forward + squash over cat(next_states, states) — both
          use pre-actor-step weights (reference samples current actions
          after the critic step, but actor weights are unchanged until the
          actor step); backward runs only on the grad-carrying states half;
        - target-critic TD forward overlaps the local-critic loss forward
          on a side stream (independent weight sets, both pre-critic-step);
        - fused loss kernels, grouped twin-Q GEMMs, one fused Adam launch
          for actor+alpha."""
        from ..ops import native
        states = batch["states"]
        actions = batch["actions"]
        rewards = batch["rewards"]
        next_states = batch["next_states"]
        dones = batch["dones"]
        T = self.num_tasks
        use_w = self.use_weighted_loss
        B = states.shape[0]
        A = self.cfg.action_dim

        import os as _os
        if getattr(self, "_bf16", False) \
                and _os.environ.get("DSAC_NO_MANUAL", "0") != "1":
            return self._update_tensors_manual(batch)
        self.zero_grad()

        # --- batched actor forward: [next | current] halves (both use
        # pre-actor-step weights; backward slices to the states half).
        # A side-stream overlap of the TD forward was A/B-tested and LOST
        # (cross-stream event sync in captured graphs cost ~5-25% —
        # profiles/r04_NOTES.md), so the TD block stays on the main stream.
        x_cat = torch.cat([next_states, states], dim=0)
        ws, bs = self._actor_weights()
        if getattr(self, "_bf16", False):
            out = Fops.mlp_forward_bf16(x_cat, ws, bs, self._actor_ws_bf16,
                                        grad_row_start=B)
        else:
            out = Fops.mlp_forward(x_cat, ws, bs, grad_row_start=B)
        mu, lsr = out[:, :A], out[:, A:]
        if self._eps_queue:
            eps = torch.cat([self._next_eps(mu[:B]), self._next_eps(mu[:B])])
        else:
            eps = torch.randn_like(mu)
        a_cat, lp_cat, ls_cat = Fops.squashed_gaussian(mu, lsr, eps,
                                                       self.actor.k)
        next_actions = a_cat[:B].detach()
        next_log_probs = lp_cat[:B].detach()
        sampled_actions = a_cat[B:]
        log_probs = lp_cat[B:]
        log_stds = ls_cat[B:]

        with torch.no_grad():
            q1_t, q2_t = self._target_q(next_states, next_actions)
            y = native().td_target_mt(
                rewards, dones, q1_t, q2_t, next_log_probs, states,
                self.log_alpha.detach(), T, self.gamma, self.reward_scale)
        q1, q2 = self._critic_q(states, actions)
        l1, l2 = Fops.critic_loss(q1, q2, y, states, self.log_alpha.detach(),
                                  T, use_w)
        q_loss = l1 + l2
        q_loss.backward()
        if self.ddp is not None:
            self.ddp.allreduce_grad_(self.critic_group.flat_grad)
        self.critic_optimizer.step()
        self.refresh_bf16("critic")

        # --- actor/alpha step (post-critic-step critic, frozen heads) --
        xa = torch.cat([states, sampled_actions], dim=-1)
        if getattr(self, "_bf16", False):
            aq1, aq2 = Fops.twin_mlp_forward_bf16(xa,
                                                  *self._twin_local_frozen,
                                                  self._twin_local_bf16)
        else:
            aq1, aq2 = Fops.twin_mlp_forward(xa, *self._twin_local_frozen)
        policy_loss, loss_log_alpha, entropy = Fops.actor_alpha_loss(
            aq1, aq2, log_probs, log_stds, states, self.log_alpha, T, use_w,
            self.H_bar_f)
        (policy_loss + loss_log_alpha).backward()
        if self.ddp is not None:
            self.ddp.allreduce_grad_(self._aa_arena)
        FusedAdam.step_many([self.actor_optimizer,
                             self.log_alpha_optimizer])
        self.alpha = self.log_alpha.exp().detach()

        flat_polyak_(self.target_group, self.critic_group, self.tau,
                     mirror=getattr(self, "_target_bf16", None))
        return {
            "critic_loss": q_loss.detach(),
            "actor_loss": policy_loss.detach(),
            "alpha_loss": loss_log_alpha.detach(),
            "entropy": entropy.detach(),
        }

    # ------------------------------------------------------------------
    # hipGraph capture: the ENTIRE update (sample gather, forward, backward,
    # three fused Adam steps, Polyak) replays as one captured graph —
    # per-step host cost collapses to one hipGraphLaunch.
    # ------------------------------------------------------------------
    def capture(self, replay, batch_size: int, warmup_iters: int = 3,
                chunk: int = 1):
        """Capture ``chunk`` FULL updates in ONE hipGraph.  hipGraphLaunch
        costs ~9 µs/node host-side (~0.22 ms for the ~25-node update), so
        at chunk=1 the host launch is nearly at parity with the 0.31 ms
        GPU time; chunking amortizes it (tools/probe_graph_k.py).  Every
        captured update is complete and distinct: the device-side RNG
        counter and Adam states advance INSIDE the graph, so update i of
        a chunk samples different replay indices and different eps.  The
        metrics dict reflects the chunk's LAST update."""
        if self._use_krng and hasattr(replay, "attach_rng"):
            replay.attach_rng(self._rng_ctr)
        assert self.device.type == "cuda", "capture needs a GPU"
        self._graph_chunk = max(1, int(chunk))
        torch.cuda.synchronize(self.device)
        side = torch.cuda.Stream(self.device)
        side.wait_stream(torch.cuda.current_stream(self.device))
        with torch.cuda.stream(side):
            for _ in range(warmup_iters):
                self.update_tensors(replay.sample(batch_size, graph_safe=True))
        torch.cuda.current_stream(self.device).wait_stream(side)
        torch.cuda.synchronize(self.device)
        graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(graph):
            for _ in range(self._graph_chunk):
                self._graph_metrics = self.update_tensors(
                    replay.sample(batch_size, graph_safe=True))
        self._graph = graph
        return graph

    def graphed_update(self) -> Dict[str, torch.Tensor]:
        """Replay the captured chunk (tensor metrics refresh in place);
        advances ``update_iteration`` by the captured chunk size."""
        self._graph.replay()
        self.update_iteration += getattr(self, "_graph_chunk", 1)
        return self._graph_metrics

    # ------------------------------------------------------------------
    # Data-parallel segmented capture: RCCL collectives are NOT
    # hipGraph-capturable on this ROCm stack (tools/probe_rccl_graph.py —
    # the capture probe aborts the process via the NCCL watchdog), so a DP
    # update is captured as THREE graphs split at the all-reduce
    # boundaries; replay runs g1 -> eager AR(critic grad) -> g2 -> eager
    # AR(actor+alpha arena) -> g3.  This recovers hipGraph launch
    # efficiency for ~all compute kernels while keeping the collectives
    # eager.  All graphs share one memory pool so cross-segment
    # intermediates (activations, squash state) keep stable addresses.
    # ------------------------------------------------------------------
    def capture_dp(self, replay, batch_size: int, warmup_iters: int = 3):
        if self._use_krng and hasattr(replay, "attach_rng"):
            replay.attach_rng(self._rng_ctr)
        assert self.device.type == "cuda", "capture needs a GPU"
        assert getattr(self, "_bf16", False), \
            "segmented DP capture runs the bf16 manual-backward path"
        torch.cuda.synchronize(self.device)
        side = torch.cuda.Stream(self.device)
        side.wait_stream(torch.cuda.current_stream(self.device))
        with torch.cuda.stream(side):
            for _ in range(warmup_iters):
                # full eager DP update (including real all-reduces: warms
                # RCCL's communicator and keeps rank launch order aligned)
                self._update_tensors_manual(
                    replay.sample(batch_size, graph_safe=True))
        torch.cuda.current_stream(self.device).wait_stream(side)
        torch.cuda.synchronize(self.device)
        g1 = torch.cuda.CUDAGraph()
        with torch.cuda.graph(g1):
            self._manual_seg1(replay.sample(batch_size, graph_safe=True))
        g2 = torch.cuda.CUDAGraph()
        with torch.cuda.graph(g2, pool=g1.pool()):
            self._manual_seg2()
        g3 = torch.cuda.CUDAGraph()
        with torch.cuda.graph(g3, pool=g1.pool()):
            self._dp_graph_metrics = self._manual_seg3()
        self._dp_graphs = (g1, g2, g3)
        return self._dp_graphs

    def dp_graphed_update(self) -> Dict[str, torch.Tensor]:
        """Replay the segmented DP update with eager RCCL between."""
        g1, g2, g3 = self._dp_graphs
        g1.replay()
        if self.ddp is not None:
            self.ddp.allreduce_grad_(self.critic_group.flat_grad)
        g2.replay()
        if self.ddp is not None:
            self.ddp.allreduce_grad_(self._aa_arena)
        g3.replay()
        self.update_iteration += 1
        return self._dp_graph_metrics

    # ------------------------------------------------------------------
    # Checkpointing — reference .tar schema (learner.save_checkpoint).
    # ------------------------------------------------------------------
    def checkpoint_state(self) -> Dict:
        def cpu_sd(m):
            return {k: v.cpu() for k, v in m.state_dict().items()}

        state = {
            "update_iteration": self.update_iteration,
            "total_step": self.total_step,
            "actor": cpu_sd(self.actor),
            "actor_optimizer": self.actor_optimizer.state_dict(),
            "critic_optimizer": self.critic_optimizer.state_dict(),
            "log_alpha": self.log_alpha.detach().cpu(),
            "log_alpha_optimizer": self.log_alpha_optimizer.state_dict(),
            "alpha": self.log_alpha.detach().exp().cpu(),
        }
        if self.variant in ("sac", "vsac"):
            # LunarLander…/src/learner.py:144-163 key layout (LL names the
            # counter 'episode_idx'; keep both for exact-schema readers)
            state["episode_idx"] = self.update_iteration
            state.update({
                "local_critic_1": cpu_sd(self.local_critic_1),
                "local_critic_2": cpu_sd(self.local_critic_2),
                "target_critic_1": cpu_sd(self.target_critic_1),
                "target_critic_2": cpu_sd(self.target_critic_2),
            })
        else:
            # MT10_Distributed_MTSAC/src/learner.py:157-174 key layout
            state.update({
                "local_critic": cpu_sd(self.local_critic),
                "target_critic": cpu_sd(self.target_critic),
            })
        return state

    def load_checkpoint_state(self, ckpt: Dict) -> None:
        """Inverse of checkpoint_state — also fixes the reference's broken
        learner resume (it referenced a nonexistent ``actor.optimizer``,
        LunarLander…/src/learner.py:178)."""
        # LL variant saves 'episode_idx' where MT variants save
        # 'update_iteration' (LunarLander…/src/learner.py:144-163)
        self.update_iteration = int(ckpt.get("update_iteration",
                                             ckpt.get("episode_idx", 0)))
        self.total_step = int(ckpt.get("total_step", 0))
        self.actor.load_state_dict(ckpt["actor"])
        if self.variant in ("sac", "vsac"):
            self.local_critic_1.load_state_dict(ckpt["local_critic_1"])
            self.local_critic_2.load_state_dict(ckpt["local_critic_2"])
            self.target_critic_1.load_state_dict(ckpt["target_critic_1"])
            self.target_critic_2.load_state_dict(ckpt["target_critic_2"])
        else:
            self.local_critic.load_state_dict(ckpt["local_critic"])
            self.target_critic.load_state_dict(ckpt["target_critic"])
        with torch.no_grad():
            self.log_alpha.copy_(ckpt["log_alpha"].to(self.device))
        self.alpha = self.log_alpha.exp().detach()
        if "actor_optimizer" in ckpt:
            self.actor_optimizer.load_state_dict(ckpt["actor_optimizer"])
        if "critic_optimizer" in ckpt:
            self.critic_optimizer.load_state_dict(ckpt["critic_optimizer"])
        if "log_alpha_optimizer" in ckpt:
            self.log_alpha_optimizer.load_state_dict(ckpt["log_alpha_optimizer"])
        self.refresh_bf16()  # bf16 compute mirrors must track loaded masters

