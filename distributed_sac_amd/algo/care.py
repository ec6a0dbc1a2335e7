"""CAREEngine — the CARE / CARE(M) learner update.

Implements the reference CARE learner exactly (MT10_Distributed_CARE/src/
learner.py:281-404, MT1_Distributed_CARE/src/learner.py):

- z_context computed once per update from the context encoder (:290);
- critic loss backward carries the ONLY gradients into the context encoder
  (actor uses z_context.detach(), :326-335), applied by a separate Adam
  after the SAC update (:399);
- the actor optimizes only its mu/log_std head (:143-149) and receives the
  critic's state encoder by hard copy every update (tau=1.0 tie, :402);
- detach_z_encs=True in the actor step stops gradients into the mixture
  (:326-335);
- target Q heads Polyak at tau, target state encoder at state_encoder_tau
  (:361-367; hardcoded 0.05 in MT1: MT1…/src/learner.py:311);
- CARE(M) (use_modified_care): frozen raw 768-d embeddings, projection in
  stateEncoder.mlp_context, weighted losses.

MI355X notes: the critic flat buffer is ordered [state-encoder params |
interleaved Q-head pairs], so the two Polyak taus are two fused kernels
over flat slices, the actor-SE tie is one flat copy, and the Q heads run
as grouped twin GEMMs.  In the GPU path, forward passes whose gradients
the reference computes but discards (actor-side trunk, critic-side encoder
in the actor step) run under no_grad — numerically identical updates.
"""

from __future__ import annotations

from typing import Dict, Optional

import torch
import torch.nn as nn

from ..config import SACConfig
from ..models.care import CAREActor, CARECritic
from ..models.context_encoder import contextEncoder
from ..ops import functional as Fops
from ..ops.flat import FlatParams, FusedAdam
from .sac import SACEngine


class CAREEngine(SACEngine):
    def __init__(self, cfg: SACConfig, device="cpu",
                 precision: Optional[str] = None):
        assert cfg.variant == "care" and cfg.encoder is not None
        super().__init__(cfg, device, precision=precision)

    # ------------------------------------------------------------------
    def _build_models(self) -> None:
        cfg, dev = self.cfg, self.device
        enc_cfg = dict(cfg.encoder)
        enc_cfg.setdefault("RoBERTa_embedding_dim", 768)
        self.enc_cfg = enc_cfg
        self.use_modified_care = cfg.use_modified_care
        self.se_tau = float(enc_cfg.get("state_encoder_tau",
                                        cfg.state_encoder_tau))
        actor_cfg = {
            "state_dim": cfg.state_dim, "action_dim": cfg.action_dim,
            "action_bound": cfg.action_bound,
            "actor_hidden_dim": cfg.actor_hidden_dim,
        }
        critic_cfg = {
            "state_dim": cfg.state_dim, "action_dim": cfg.action_dim,
            "critic_hidden_dim": cfg.critic_hidden_dim,
        }
        self.context_encoder = contextEncoder(enc_cfg,
                                              self.use_modified_care).to(dev)
        self.actor = CAREActor(actor_cfg, enc_cfg, self.use_modified_care).to(dev)
        self.local_critic = CARECritic(critic_cfg, enc_cfg,
                                       self.use_modified_care).to(dev)
        self.target_critic = CARECritic(critic_cfg, enc_cfg,
                                        self.use_modified_care).to(dev)
        self.log_alpha = nn.Parameter(torch.full(
            (cfg.num_tasks,), float(cfg.log_alpha), device=dev))
        self.H_bar = torch.tensor([-float(cfg.action_dim)], device=dev)
        self.H_bar_f = -float(cfg.action_dim)
        self.alpha = self.log_alpha.exp().detach()
        # CARE uses weighted losses iff modified CARE
        # (reference learner.update_SAC use_weighted_loss=use_modified_care)
        self.use_weighted_loss = self.use_modified_care
        # strict-parity degenerate-broadcast mode (docs/PARITY.md)
        self.degenerate_w = \
            getattr(cfg, "weighted_loss_mode", "corrected") == "reference"

    def _critic_layer_pairs(self, target: bool = False):
        c = self.target_critic if target else self.local_critic
        return self._critic_linears(c)

    def _se_then_heads(self, critic: CARECritic):
        order = list(critic.state_encoder.parameters())
        self._se_numel = sum(p.numel() for p in order)
        for l1, l2 in self._critic_linears(critic):
            order += [l1.weight, l2.weight, l1.bias, l2.bias]
        return order

    def _build_optimizers(self) -> None:
        cfg = self.cfg
        self.actor_group = FlatParams(self.actor.mu_log_std_layer.parameters())
        self.actor_se_group = FlatParams(self.actor.state_encoder.parameters(),
                                         with_grad=False)
        self.critic_group = FlatParams(self._se_then_heads(self.local_critic))
        self.target_group = FlatParams(self._se_then_heads(self.target_critic),
                                       with_grad=False)
        self.alpha_group = FlatParams([self.log_alpha])
        ctx_params = [p for p in self.context_encoder.parameters()
                      if p.requires_grad]
        self.context_group = FlatParams(ctx_params) if ctx_params else None

        self.actor_optimizer = FusedAdam(self.actor_group, lr=cfg.lr_actor)
        self.critic_optimizer = FusedAdam(
            self.critic_group, lr=cfg.lr_critic,
            ref_params=list(self.local_critic.parameters()))
        self.log_alpha_optimizer = FusedAdam(self.alpha_group, lr=cfg.lr_actor)
        self.context_encoder_optimizer = (
            FusedAdam(self.context_group,
                      lr=float(self.enc_cfg.get("lr_contextEnc", cfg.lr_actor)))
            if self.context_group is not None else None)

        self._twin_local = self._build_twin_stacks(
            self.critic_group, self._critic_layer_pairs(), with_grad=True)
        self._twin_target = self._build_twin_stacks(
            self.target_group, self._critic_layer_pairs(target=True),
            with_grad=False)
        self._twin_local_frozen = ([w.detach() for w in self._twin_local[0]],
                                   [b.detach() for b in self._twin_local[1]])
        self._aa_arena = torch.zeros(
            self.actor_group.numel + self.alpha_group.numel,
            device=self.device)
        off = self.actor_group.adopt_grad_arena(self._aa_arena, 0)
        self.alpha_group.adopt_grad_arena(self._aa_arena, off)
        self._init_bf16_mirrors()
        self._init_care_bf16()
        self.hard_copy_targets()
        self.tie_actor_state_encoder()

    def attach_ddp(self, ddp) -> None:
        self.ddp = ddp
        if ddp is not None and ddp.enabled:
            for g in (self.actor_group, self.critic_group, self.alpha_group,
                      self.actor_se_group):
                ddp.broadcast_params(g.flat_data)
            if self.context_group is not None:
                ddp.broadcast_params(self.context_group.flat_data)
            self.hard_copy_targets()
            self.refresh_bf16()  # mirrors must track broadcast params

    @torch.no_grad()
    def tie_actor_state_encoder(self) -> None:
        """actor.state_encoder <- critic.state_encoder hard copy (reference
        soft_update(..., tau=1.0), learner.py:402) as one flat copy."""
        self.actor_se_group.flat_data.copy_(
            self.critic_group.flat_data[: self._se_numel])

    def zero_grad(self) -> None:
        self.actor_group.zero_grad()
        self.critic_group.zero_grad()
        self.alpha_group.zero_grad()
        if self.context_group is not None:
            self.context_group.zero_grad()
        self.actor_group.rebind_grads()
        self.critic_group.rebind_grads()
        self.alpha_group.rebind_grads()
        if self.context_group is not None:
            self.context_group.rebind_grads()

    def _polyak_targets(self, mirror=None) -> None:
        """Q heads at tau, state encoder at state_encoder_tau — two fused
        kernels over the flat slices (reference learner.py:361-367); with
        ``mirror`` the target bf16 mirror slices refresh in-kernel."""
        se = self._se_numel
        t, s = self.target_group.flat_data, self.critic_group.flat_data
        m0 = mirror[:se] if mirror is not None else None
        m1 = mirror[se:] if mirror is not None else None
        self._polyak_slice(t[:se], s[:se], self.se_tau, m0)
        self._polyak_slice(t[se:], s[se:], self.tau, m1)

    @staticmethod
    @torch.no_grad()
    def _polyak_slice(t: torch.Tensor, s: torch.Tensor, tau: float,
                      mirror=None) -> None:
        from ..ops import has_native, native, native_enabled
        if t.is_cuda and native_enabled() and has_native():
            if mirror is not None:
                native().polyak_(t, s, float(tau), mirror)
            else:
                native().polyak_(t, s, float(tau))
            return
        t.mul_(1.0 - tau).add_(s, alpha=tau)
        if mirror is not None:
            mirror.copy_(t)

    # ------------------------------------------------------------------
    def _sample_care(self, mtobss, z_context, detach_z_encs=False):
        mu, log_std_raw = self.actor(mtobss, z_context, detach_z_encs)
        eps = self._next_eps(mu)
        return Fops.squashed_gaussian(mu, log_std_raw, eps, self.actor.k)

    def update_tensors(self, batch: Dict[str, torch.Tensor]):
        if self._use_fused(batch["states"]):
            if self.degenerate_w and self.use_weighted_loss:
                raise RuntimeError(
                    "weighted_loss_mode='reference' is a CPU strict-parity "
                    "mode; the fused HIP kernels implement corrected "
                    "weighting (docs/PARITY.md)")
            return self._update_tensors_fused(batch)
        states, actions = batch["states"], batch["actions"]
        rewards, next_states, dones = (batch["rewards"], batch["next_states"],
                                       batch["dones"])
        T = self.num_tasks
        alpha = Fops.gather_log_alpha(states[:, -T:],
                                      self.log_alpha).exp().detach()
        self.zero_grad()

        # z_context once per update (reference :290)
        z_context = self.context_encoder(states)
        with torch.no_grad():
            na, nlp, _ = self._sample_care(next_states, z_context)
            q1_t, q2_t = self.target_critic(next_states, z_context, na)
            y = Fops.td_target(rewards, dones, q1_t, q2_t, nlp, alpha,
                               self.gamma, self.reward_scale)

        l1, l2 = self.local_critic.cal_loss(
            states, z_context, actions, y,
            use_weighted_loss=self.use_weighted_loss, num_tasks=T,
            alphas=self.log_alpha.exp().detach(),
            degenerate=self.degenerate_w)
        q_loss = l1 + l2
        q_loss.backward(retain_graph=True)
        if self.ddp is not None:
            self.ddp.allreduce_grad_(self.critic_group.flat_grad)
        self.critic_optimizer.step()

        zc_d = z_context.detach()
        sa, lp, ls = self._sample_care(states, zc_d, detach_z_encs=True)
        q1, q2 = self.local_critic(states, zc_d, sa, detach_z_encs=True)
        q_min = torch.min(q1, q2)
        policy_loss = self.actor.cal_loss(
            lp, q_min, alpha, use_weighted_loss=self.use_weighted_loss,
            mtobss=states, num_tasks=T,
            alphas=self.log_alpha.exp().detach(),
            degenerate=self.degenerate_w)
        policy_loss.backward()
        if self.ddp is not None:
            self.ddp.allreduce_grad_(self.actor_group.flat_grad)
        self.actor_optimizer.step()

        entropy = Fops.entropy_from_log_std(ls)
        log_alpha_g = Fops.gather_log_alpha(states[:, -T:], self.log_alpha)
        loss_log_alpha = -(log_alpha_g * (lp.detach() + self.H_bar)).mean()
        loss_log_alpha.backward()
        if self.ddp is not None:
            self.ddp.allreduce_grad_(self.alpha_group.flat_grad)
        self.log_alpha_optimizer.step()
        self.alpha = self.log_alpha.exp().detach()

        self._polyak_targets()
        if self.context_encoder_optimizer is not None:
            if self.ddp is not None:
                self.ddp.allreduce_grad_(self.context_group.flat_grad)
            self.context_encoder_optimizer.step()   # grads from critic bwd
        self.tie_actor_state_encoder()
        return {"critic_loss": q_loss.detach(),
                "actor_loss": policy_loss.detach(),
                "alpha_loss": loss_log_alpha.detach(),
                "entropy": entropy.detach()}

    # -- bf16 fast state-encoder machinery ------------------------------
    def _init_care_bf16(self) -> None:
        if not getattr(self, "_bf16", False):
            self._se_fast = False
            return
        self._se_fast = True
        dev = self.device

        def mlp_linears(seq):
            import torch.nn as nn_
            return [m for m in seq if isinstance(m, nn_.Linear)]

        def build(critic, group, mirror):
            se = critic.state_encoder
            mix = [m for m in se.mixture_encoders.mixtureEncoders
                   if hasattr(m, "W")]
            k = mix[0].num_encoders
            info = {"k": k,
                    "mixW": [m.W for m in mix],
                    "mixB": [m.b for m in mix],
                    "mixT": [torch.empty(k, m.W.shape[2], m.W.shape[1],
                                         dtype=torch.bfloat16, device=dev)
                             for m in mix]}
            for name, seq in (("trunk", se.trunk),
                              ("mlpctx", getattr(se, "mlp_context", None))):
                if seq is None:
                    info[name] = None
                    continue
                lins = mlp_linears(seq)
                ws, bs, wsh = [], [], []
                for l in lins:
                    i = next(j for j, q in enumerate(group.params)
                             if q is l.weight)
                    off = group.offsets[i]
                    wsh.append(mirror[off:off + l.weight.numel()]
                               .view_as(l.weight))
                    ws.append(l.weight)
                    bs.append(l.bias)
                info[name] = (ws, bs, wsh)
            # fused narrow-chain eligibility: every layer <= 64 wide,
            # <= 6 layers (k_bf16_mlp_narrow constraints)
            def narrow_ok(widths, L):
                return L <= 6 and all(w <= 64 for w in widths)
            info["mix_narrow"] = narrow_ok(
                [m.W.shape[2] for m in mix], len(mix))
            info["trunk_narrow"] = narrow_ok(
                [w.shape[0] for w in info["trunk"][0]],
                len(info["trunk"][0]))
            info["ctx_narrow"] = (info["mlpctx"] is not None and narrow_ok(
                [w.shape[0] for w in info["mlpctx"][0]],
                len(info["mlpctx"][0])))
            return info

        self._se_local = build(self.local_critic, self.critic_group,
                               self._critic_bf16)
        self._se_target = build(self.target_critic, self.target_group,
                                self._target_bf16)
        self._refresh_mixT()

        # original-CARE trainable context encoder (round 2): bf16 mirror
        # + W^T buffers so the manual path can run the context chain on
        # the fused chain kernels (fwd + dx + grouped dwdb, third arena)
        self._ctx_chain = None
        if self.context_group is not None:
            import torch.nn as nn_
            lins = [m for m in self.context_encoder.modules()
                    if isinstance(m, nn_.Linear)]
            grp = self.context_group
            mir = torch.empty(grp.numel, dtype=torch.bfloat16, device=dev)
            ws16, wts, bss = [], [], []
            for l in lins:
                i = next(j for j, q in enumerate(grp.params)
                         if q is l.weight)
                off = grp.offsets[i]
                ws16.append(mir[off:off + l.weight.numel()]
                            .view_as(l.weight))
                wts.append(torch.empty(l.weight.shape[1],
                                       l.weight.shape[0],
                                       dtype=torch.bfloat16, device=dev))
                bss.append(l.bias)
            mir.copy_(grp.flat_data)
            self.context_encoder_optimizer.bf16_mirror = mir

            def _fpx(w, dx):
                K = w.shape[-1]
                N = w.numel() // K
                r, c = (K, N) if dx else (N, K)
                return torch.empty(
                    ((r + 15) // 16) * ((c + 31) // 32) * 512,
                    dtype=torch.bfloat16, device=dev)
            self._ctx_chain = {"mir": mir, "ws16": ws16, "wt": wts,
                               "bs": bss, "lins": lins,
                               "fp": [_fpx(w, False) for w in ws16],
                               "dxp": [_fpx(w, True) for w in ws16]}

    @torch.no_grad()
    def _refresh_mixT(self, which: str = "all") -> None:
        if not getattr(self, "_se_fast", False):
            return
        if which in ("all", "critic"):
            for m, W in zip(self._se_local["mixT"], self._se_local["mixW"]):
                m.copy_(W.detach().permute(0, 2, 1))
        if which in ("all", "target"):
            for m, W in zip(self._se_target["mixT"],
                            self._se_target["mixW"]):
                m.copy_(W.detach().permute(0, 2, 1))

    def refresh_bf16(self, which: str = "all") -> None:
        super().refresh_bf16(which)
        self._refresh_mixT(which)
        if which == "all" and getattr(self, "_ctx_chain", None) is not None:
            self._ctx_chain["mir"].copy_(self.context_group.flat_data)

    def _se_fwd_fast(self, info, mtobss_2d, z_context):
        """stateEncoder.forward via bf16 kernels (value-identical to the
        module path up to bf16 rounding; skips the reference's divide by
        alpha.sum(dim=1) which is exactly softmax-sum==1)."""
        # mixture encoders consume the RAW state — strip the one-hot
        # suffix (reference stateEncoder.mtobss2states_taskIndices)
        states_2d = mtobss_2d[:, : mtobss_2d.shape[1] - self.num_tasks]
        z_encs = Fops.grouped_mlp_bf16(states_2d, info["mixW"],
                                       info["mixB"], info["mixT"],
                                       info["k"])              # [k,B,50] f32
        tws, tbs, twsh = info["trunk"]
        logits = Fops.mlp_forward_bf16(z_context.detach(), tws, tbs, twsh)
        alpha = torch.softmax(logits, dim=-1)                  # [B,k]
        z_enc = (z_encs * alpha.t().unsqueeze(-1)).sum(0)      # [B,50]
        if info["mlpctx"] is not None:
            cws, cbs, cwsh = info["mlpctx"]
            zc = Fops.mlp_forward_bf16(z_context, cws, cbs, cwsh)
        else:
            zc = z_context
        return torch.cat([zc, z_enc], dim=1)

    def _twin_fwd(self, x, which: str):
        if getattr(self, "_bf16", False):
            if which == "target":
                return Fops.twin_mlp_forward_bf16(x, *self._twin_target,
                                                  self._twin_target_bf16)
            if which == "frozen":
                return Fops.twin_mlp_forward_bf16(x,
                                                  *self._twin_local_frozen,
                                                  self._twin_local_bf16)
            return Fops.twin_mlp_forward_bf16(x, *self._twin_local,
                                              self._twin_local_bf16)
        if which == "target":
            return Fops.twin_mlp_forward(x, *self._twin_target)
        if which == "frozen":
            return Fops.twin_mlp_forward(x, *self._twin_local_frozen)
        return Fops.twin_mlp_forward(x, *self._twin_local)

    def _update_tensors_fused(self, batch):
        """GPU path: fused losses + twin-head grouped GEMMs (bf16 mirrors
        when enabled).  Forwards whose grads the reference discards run
        under no_grad (see module doc); the actor head runs ONE batched
        pass over cat(next, states) like the SAC engine (the actor-side
        state encoder is no-grad in both halves)."""
        import os as _os
        manual_ok = getattr(self, "_se_fast", False) and (
            self.use_modified_care
            or (getattr(self, "_ctx_chain", None) is not None
                and self._use_chain))
        if manual_ok and _os.environ.get("DSAC_NO_MANUAL", "0") != "1":
            return self._update_tensors_manual(batch)
        from ..ops import native
        states, actions = batch["states"], batch["actions"]
        rewards, next_states, dones = (batch["rewards"], batch["next_states"],
                                       batch["dones"])
        T = self.num_tasks
        use_w = self.use_weighted_loss
        B = states.shape[0]
        A = self.actor.action_dim
        self.zero_grad()

        z_context = self.context_encoder(states)
        zc_d = z_context.detach()

        # ONE batched local-SE forward over [next | current]: the actor's
        # tied encoder equals the pre-step critic encoder (hard copy at the
        # end of every update), so its encodings are the same VALUES — the
        # states half doubles as the critic-loss encoding (with grad), and
        # a detached copy feeds the actor head.
        x_cat = torch.cat([next_states, states], dim=0)
        if getattr(self, "_se_fast", False):
            z2 = torch.cat([z_context, z_context], dim=0)
            enc_cat = self._se_fwd_fast(self._se_local, x_cat, z2)
        else:
            with torch.no_grad():
                z2 = torch.cat([zc_d, zc_d], dim=0)
                enc_cat = self.actor.state_encoder(z2, x_cat,
                                                   detach_z_encs=True)
        ws, bs = self._actor_weights()
        enc_actor_in = (enc_cat.detach() if getattr(self, "_se_fast", False)
                        else enc_cat)
        if getattr(self, "_bf16", False):
            mu_lsr = Fops.mlp_forward_bf16(enc_actor_in, ws, bs,
                                           self._actor_ws_bf16,
                                           grad_row_start=B)
        else:
            mu_lsr = Fops.mlp_forward(enc_actor_in, ws, bs, grad_row_start=B)
        mu = mu_lsr[:, :A]
        lsr = mu_lsr[:, A:]
        if self._eps_queue:
            eps = torch.cat([self._next_eps(mu[:B]), self._next_eps(mu[:B])])
        else:
            eps = torch.randn_like(mu)
        a_cat, lp_cat, ls_cat = Fops.squashed_gaussian(mu, lsr, eps,
                                                       self.actor.k)
        na, nlp = a_cat[:B].detach(), lp_cat[:B].detach()
        sa, lp, ls = a_cat[B:], lp_cat[B:], ls_cat[B:]

        with torch.no_grad():
            if getattr(self, "_se_fast", False):
                enc_t = self._se_fwd_fast(self._se_target, next_states,
                                          z_context)
            else:
                enc_t = self.target_critic.encode(next_states, z_context)
            xt = torch.cat([enc_t, na], dim=-1)
            q1_t, q2_t = self._twin_fwd(xt, "target")
            y = native().td_target_mt(rewards, dones, q1_t, q2_t, nlp,
                                      states, self.log_alpha.detach(), T,
                                      self.gamma, self.reward_scale)

        if getattr(self, "_se_fast", False):
            enc = enc_cat[B:]
        else:
            enc = self.local_critic.encode(states, z_context)
        x = torch.cat([enc, actions], dim=-1)
        q1, q2 = self._twin_fwd(x, "local")
        l1, l2 = Fops.critic_loss(q1, q2, y, states, self.log_alpha.detach(),
                                  T, use_w)
        q_loss = l1 + l2
        q_loss.backward()
        if self.ddp is not None:
            self.ddp.allreduce_grad_(self.critic_group.flat_grad)
            if self.context_group is not None:
                self.ddp.allreduce_grad_(self.context_group.flat_grad)
        self.critic_optimizer.step()
        self.refresh_bf16("critic")

        with torch.no_grad():
            if getattr(self, "_se_fast", False):
                enc_c = self._se_fwd_fast(self._se_local, states, zc_d)
            else:
                enc_c = self.local_critic.encode(states, zc_d,
                                                 detach_z_encs=True)
        xa = torch.cat([enc_c, sa], dim=-1)
        aq1, aq2 = self._twin_fwd(xa, "frozen")
        policy_loss, loss_log_alpha, entropy = Fops.actor_alpha_loss(
            aq1, aq2, lp, ls, states, self.log_alpha, T, use_w, self.H_bar_f)
        (policy_loss + loss_log_alpha).backward()
        if self.ddp is not None:
            self.ddp.allreduce_grad_(self._aa_arena)
        from ..ops.flat import FusedAdam as _FA
        _FA.step_many([self.actor_optimizer, self.log_alpha_optimizer])
        self.refresh_bf16("actor")
        self.alpha = self.log_alpha.exp().detach()

        self._polyak_targets()
        self.refresh_bf16("target")
        if self.context_encoder_optimizer is not None:
            self.context_encoder_optimizer.step()
        self.tie_actor_state_encoder()
        return {"critic_loss": q_loss.detach(),
                "actor_loss": policy_loss.detach(),
                "alpha_loss": loss_log_alpha.detach(),
                "entropy": entropy.detach()}


    # -- hand-rolled backward (modified CARE, bf16) ----------------------
    @torch.no_grad()
    def _se_fwd_manual(self, info, states_bf16, zc16, save=False, rep=1):
        """stateEncoder forward on raw kernels, optionally saving the
        activations the manual backward needs.  Value-identical to
        :meth:`_se_fwd_fast` up to bf16 rounding (the attention pool runs
        in one fused kernel, fp32 accumulate).

        ``rep=2``: the batched [next | current] call — ``zc16`` carries
        ONE copy of z_context (B rows) and the trunk / context
        projections run once on it (round-1 computed them on duplicated
        2B rows); the fused pool indexes alpha/hc by ``row % B`` and
        writes the CONCATENATED head input directly (no torch.cat)."""
        from ..ops import native
        ext = native()
        k = info["k"]
        sv = 1 if save else 0
        tws, tbs, twsh = info["trunk"]
        all_narrow = (info["mix_narrow"] and info["trunk_narrow"]
                      and (info["mlpctx"] is None or info["ctx_narrow"]))
        if all_narrow:
            # round 2: the whole SE forward's independent chains
            # (mixture G=k, trunk, mlp_context) run in ONE launch
            xs = [states_bf16, zc16]
            wss = [list(info["mixT"]), list(twsh)]
            bss = [list(info["mixB"]), [b.contiguous() for b in tbs]]
            Gs, als, ofs = [k, 1], [0, 0], [1, 1]
            if info["mlpctx"] is not None:
                cws_, cbs_, cwsh_ = info["mlpctx"]
                xs.append(zc16)
                wss.append(list(cwsh_))
                bss.append([b.contiguous() for b in cbs_])
                Gs.append(1)
                als.append(0)
                ofs.append(0)
            res = ext.mlp_narrow_fwd_multi(xs, wss, bss, Gs, als, ofs, sv)
            z_encs = res[0][0]
            acts_m = [states_bf16] + list(res[0][1:])
            h = res[1][0]
            acts_t = [zc16] + list(res[1][1:])
            if info["mlpctx"] is not None:
                hc = res[2][0]
                acts_c = [zc16] + list(res[2][1:])
                alpha, enc = ext.attn_pool_fwd_enc(h, z_encs, hc, rep)
                if save:
                    return enc, dict(acts_m=acts_m, z_encs=z_encs,
                                     acts_t=acts_t, alpha=alpha,
                                     acts_c=acts_c, rep=rep)
                return enc, None
            alpha, enc = ext.attn_pool_fwd_enc(h, z_encs, zc16, rep)
            if save:
                return enc, dict(acts_m=acts_m, z_encs=z_encs,
                                 acts_t=acts_t, alpha=alpha, acts_c=None,
                                 rep=rep)
            return enc, None
        if info["mix_narrow"]:
            res = ext.mlp_narrow_fwd_bf16(states_bf16, info["mixT"],
                                          list(info["mixB"]), k, 0, 1, sv)
            z_encs = res[0]                      # f32 [k,M,D]
            acts_m = [states_bf16] + res[1:]
        else:
            acts_m = [states_bf16]
            h = states_bf16
            nm = len(info["mixT"])
            z_encs = None
            for i in range(nm):
                last = i == nm - 1
                h = ext.linear_act_fwd_bf16(h, info["mixT"][i],
                                            info["mixB"][i],
                                            0 if last else 1, k,
                                            1 if last else 0)
                if last:
                    z_encs = h
                else:
                    acts_m.append(h)
        if info["trunk_narrow"]:
            res = ext.mlp_narrow_fwd_bf16(zc16, twsh, list(tbs), 1, 0, 1, sv)
            h = res[0]                           # logits f32
            acts_t = [zc16] + res[1:]
        else:
            acts_t = [zc16]
            h = zc16
            nt = len(twsh)
            for i in range(nt):
                last = i == nt - 1
                h = ext.linear_act_fwd_bf16(h, twsh[i], tbs[i].contiguous(),
                                            0 if last else 1, 1,
                                            1 if last else 0)
                if not last:
                    acts_t.append(h)
        if info["mlpctx"] is None:
            # original CARE: no mlp_context — enc = [z_context | z_enc]
            alpha, enc = ext.attn_pool_fwd_enc(h, z_encs, zc16, rep)
            if save:
                return enc, dict(acts_m=acts_m, z_encs=z_encs,
                                 acts_t=acts_t, alpha=alpha, acts_c=None,
                                 rep=rep)
            return enc, None
        cws, cbs, cwsh = info["mlpctx"]
        if info["ctx_narrow"]:
            res = ext.mlp_narrow_fwd_bf16(zc16, cwsh, list(cbs), 1, 0, 0, sv)
            hc = res[0]                          # zc projection, bf16
            acts_c = [zc16] + res[1:]
        else:
            acts_c = [zc16]
            hc = zc16
            nc = len(cwsh)
            for i in range(nc):
                last = i == nc - 1
                hc = ext.linear_act_fwd_bf16(hc, cwsh[i],
                                             cbs[i].contiguous(),
                                             0 if last else 1, 1, 0)
                if not last:
                    acts_c.append(hc)
        alpha, enc = ext.attn_pool_fwd_enc(h, z_encs, hc, rep)
        if save:
            return enc, dict(acts_m=acts_m, z_encs=z_encs, acts_t=acts_t,
                             alpha=alpha, acts_c=acts_c, rep=rep)
        return enc, None

    @torch.no_grad()
    def _update_tensors_manual(self, batch):
        """CARE bf16 update with a HAND-ROLLED backward (both variants:
        CARE(M) frozen context, original CARE trainable context encoder),
        structured as three SEGMENTS split at the DP all-reduce
        boundaries so capture_dp (inherited from SACEngine) can hipGraph
        each segment with eager RCCL between replays:
          seg1: z_context + SE/actor forwards + TD + critic(+context)
                backward -> critic (+ context) grads
          seg2: critic Adam + actor-side forward/backward -> aa grads
          seg3: actor/alpha Adam + Polyak + context Adam + SE tie
        (reference gradient-flow rules: MT10_Distributed_CARE/src/
        learner.py:281-404 — trunk input detached, actor optimizes only
        its head, context grads from the critic loss only)."""
        self._manual_seg1(batch)
        if self.ddp is not None:
            self.ddp.allreduce_grad_(self.critic_group.flat_grad)
            if self._dp_st.get("orig") and self.context_group is not None:
                self.ddp.allreduce_grad_(self.context_group.flat_grad)
        self._manual_seg2()
        if self.ddp is not None:
            self.ddp.allreduce_grad_(self._aa_arena)
        return self._manual_seg3()

    def dp_graphed_update(self):
        """Segmented DP replay with CARE's extra context all-reduce."""
        g1, g2, g3 = self._dp_graphs
        g1.replay()
        if self.ddp is not None:
            self.ddp.allreduce_grad_(self.critic_group.flat_grad)
            if (not self.use_modified_care
                    and self.context_group is not None):
                self.ddp.allreduce_grad_(self.context_group.flat_grad)
        g2.replay()
        if self.ddp is not None:
            self.ddp.allreduce_grad_(self._aa_arena)
        g3.replay()
        self.update_iteration += 1
        return self._dp_graph_metrics

    @torch.no_grad()
    def _manual_seg1(self, batch):
        from ..ops import native
        ext = native()
        states, actions = batch["states"], batch["actions"]
        rewards, next_states, dones = (batch["rewards"],
                                       batch["next_states"], batch["dones"])
        T = self.num_tasks
        use_w = self.use_weighted_loss
        B = states.shape[0]
        A = self.actor.action_dim
        la_det = self.log_alpha.detach()
        sd = states.shape[1] - T                 # raw state dim
        D = self._se_local["mixW"][-1].shape[2]
        info = self._se_local
        nl_c = len(self._twin_local_bf16)
        nl_a = len(self._actor_ws_bf16)
        orig = not self.use_modified_care
        chain0 = self._use_chain
        if chain0:
            # ONE launch re-packs every pre-step weight set
            ws_pack = (list(self._twin_local_bf16) * 2
                       + list(self._twin_target_bf16)
                       + list(self._actor_ws_bf16) * 2)
            ps_pack = (list(self._twin_local_fp)
                       + list(self._twin_local_dxp)
                       + list(self._twin_target_fp)
                       + list(self._actor_fp) + list(self._actor_dxp))
            gs_pack = [2] * (3 * nl_c) + [1] * (2 * nl_a)
            dx_pack = ([0] * nl_c + [1] * nl_c + [0] * nl_c
                       + [0] * nl_a + [1] * nl_a)
            if orig:
                cc0 = self._ctx_chain
                ws_pack += list(cc0["ws16"]) * 2
                ps_pack += list(cc0["fp"]) + list(cc0["dxp"])
                gs_pack += [1] * (2 * len(cc0["ws16"]))
                dx_pack += [0] * len(cc0["ws16"]) + [1] * len(cc0["ws16"])
            ext.pack_weights_frag(ws_pack, ps_pack, gs_pack, dx_pack)

        if orig:
            # original CARE: TRAINABLE context encoder (embedding header +
            # mlp) — forward on the fused chain kernel over the gathered
            # frozen embeddings; activations saved for the third-arena
            # backward below (reference MT1…/learner.py: context grads
            # come from the critic loss only)
            cc = self._ctx_chain
            idx = states[:, -T:].argmax(dim=1)
            emb = torch.relu(self.context_encoder.embedding[0](idx))
            z_context, acts_ctx = self._chain_fwd(
                emb, None, cc["ws16"], cc["bs"], G=1, act_last=0,
                out_f32=True,
                wps=cc["fp"] if self._use_chain else None)
        else:
            z_context = self.context_encoder(states)     # frozen embedding
            acts_ctx = None
        zc_dim = (self._se_local["mlpctx"][0][-1].shape[0]
                  if self._se_local["mlpctx"] is not None
                  else z_context.shape[1])
        zc16 = z_context.to(torch.bfloat16)

        # ---- ONE batched local-SE forward over [next | current]; the
        # trunk / context projections run ONCE on the (identical) z rows
        x_cat = torch.cat([next_states[:, :sd], states[:, :sd]], dim=0)
        enc_cat, se_saved = self._se_fwd_manual(info, x_cat.to(torch.bfloat16),
                                                zc16, save=True, rep=2)

        # ---- batched actor head + squash ------------------------------
        ws_f32, bs_f32 = self._actor_weights()
        chain = self._use_chain
        if chain:
            out, acts_a = self._chain_fwd(enc_cat, None,
                                          self._actor_ws_bf16, bs_f32,
                                          G=1, wps=self._actor_fp)
        else:
            out, acts_a = self._mlp_fwd_manual(enc_cat, self._actor_ws_bf16,
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
        sa, lp = a_cat[B:], lp_cat[B:]

        # ---- TD target (target SE + target heads, no grad) ------------
        enc_t, _ = self._se_fwd_manual(self._se_target,
                                       next_states[:, :sd].to(torch.bfloat16),
                                       zc16)
        if chain:
            yt, _ = self._chain_fwd(enc_t, na, self._twin_target_bf16,
                                    self._twin_target[1], G=2,
                                    save_acts=False,
                                    wps=self._twin_target_fp)
            q1_t, q2_t = yt[0], yt[1]
        else:
            xt = torch.cat([enc_t, na.to(torch.bfloat16)], dim=-1)
            q1_t, q2_t, _ = self._twin_fwd_manual(xt,
                                                  self._twin_target_bf16,
                                                  self._twin_target[1])
        y = ext.td_target_mt(rewards, dones, q1_t, q2_t, nlp, states,
                             la_det, T, self.gamma, self.reward_scale)

        # ---- critic loss + manual backward ----------------------------
        if chain:
            yq, acts_q = self._chain_fwd(enc_cat[B:], actions,
                                         self._twin_local_bf16,
                                         self._twin_local[1], G=2,
                                         wps=self._twin_local_fp)
            q1, q2 = yq[0], yq[1]
            head_in_dim = acts_q[0].shape[-1]
        else:
            x = torch.cat([enc_cat[B:], actions.to(torch.bfloat16)], dim=-1)
            q1, q2, acts_q = self._twin_fwd_manual(x, self._twin_local_bf16,
                                                   self._twin_local[1])
            head_in_dim = x.shape[1]
        closs = ext.critic_loss_fwd(q1, q2, y, states, la_det, T,
                                    int(use_w), self._loss_ws)[0]
        dy = ext.critic_loss_bwd2(q1, q2, y, states, la_det, closs, T,
                                  int(use_w))
        wsg, bsg = self._twin_local
        fg_c = self.critic_group.flat_grad
        arena_c, S_c, ch_c = self._dw_arena("critic",
                                            self.critic_group.numel, B)
        base_c = fg_c.data_ptr()
        if chain:
            empty_h = states.new_empty(0, dtype=torch.bfloat16)
            youts_q = [acts_q[i + 1] for i in range(nl_c - 1)] + [empty_h]
            aflags_q = [1] * (nl_c - 1) + [0]
            outs = ext.mlp_chain_dx_bf16(dy, list(self._twin_local_wt),
                                         youts_q, head_in_dim, aflags_q,
                                         2, 1, 0,
                                         list(self._twin_local_dxp))
            dys_q, dx0f = outs[:-1], outs[-1]
            dx0 = (dx0f[0] + dx0f[1]).to(torch.bfloat16)  # [B, se+A]
            ext.dwdb_grouped_arena(
                list(dys_q), [acts_q[i] for i in range(nl_c)], arena_c,
                [(w.grad.data_ptr() - base_c) // 4 for w in wsg],
                [(b.grad.data_ptr() - base_c) // 4 for b in bsg],
                2, S_c, ch_c)
        else:
            for i in range(nl_c - 1, -1, -1):
                act = 1 if i < nl_c - 1 else 0
                yout = acts_q[i + 1] if i < nl_c - 1 else acts_q[i]
                ext.linear_bwd_dwdb_arena(
                    dy, acts_q[i], yout, act, 2, arena_c,
                    (wsg[i].grad.data_ptr() - base_c) // 4,
                    (bsg[i].grad.data_ptr() - base_c) // 4, S_c, ch_c, 0)
                if i > 0:
                    dy = ext.linear_bwd_dx_bf16(dy, self._twin_local_bf16[i],
                                                yout, act, 2, 0)
            dx0 = ext.linear_bwd_dx_bf16(dy, self._twin_local_bf16[0],
                                         acts_q[1] if nl_c > 1 else acts_q[0],
                                         1 if nl_c > 1 else 0, 2, 1)  # [B,se+A]
        # state-encoder backward (states half of the batched acts)
        zen = se_saved["z_encs"][:, B:].contiguous()     # f32 [k,B,D]
        dedup = se_saved.get("rep", 1) == 2
        alpha_cur = (se_saved["alpha"] if dedup
                     else se_saved["alpha"][B:])
        dzencs, dlogits = ext.attn_pool_bwd(zen, alpha_cur,
                                            dx0, dx0.shape[1], zc_dim)
        mlpctx = info["mlpctx"]
        if mlpctx is not None:
            cws, cbs, cwsh = mlpctx
        tws, tbs, twsh = info["trunk"]
        acts_m = ([se_saved["acts_m"][0][B:]]
                  + [a[:, B:].contiguous() for a in se_saved["acts_m"][1:]])
        import os as _os
        use_fused_se_bwd = (
            _os.environ.get("DSAC_NARROW_BWD", "1") == "1"
            and info["mix_narrow"] and info["trunk_narrow"]
            and (mlpctx is None or info["ctx_narrow"]))
        if use_fused_se_bwd:
            # DEFAULT ON since round 2: GPU-validated (exact parameter
            # agreement vs the per-layer arena path, +3.8% eager rate —
            # gpurun_out r2 call1 / tests/test_gpu_kernels.py narrow gate
            # test); whole-chain fused backward, one launch per chain,
            # partials per 64-row block.  DSAC_NARROW_BWD=0 restores the
            # per-layer path.
            base = fg_c.data_ptr()

            def offs(ps):
                return [(p_.grad.data_ptr() - base) // 4 for p_ in ps]
            arena2, S2, _ = self._dw_arena("se@rowblocks",
                                           self.critic_group.numel, B)
            def cur_half(acts):
                return acts if dedup else [a[B:] for a in acts]
            # all three SE backward chains in ONE launch (round 2)
            dys_l, actss, wss_l = [], [], []
            w_os, b_os, Gl, trl = [], [], [], []
            if mlpctx is not None:
                dys_l.append(dx0[:, :zc_dim].contiguous())
                actss.append(cur_half(se_saved["acts_c"]))
                wss_l.append(list(cwsh))
                w_os.append(offs(cws))
                b_os.append(offs(cbs))
                Gl.append(1)
                trl.append(0)
            dys_l.append(dlogits)
            actss.append(cur_half(se_saved["acts_t"]))
            wss_l.append(list(twsh))
            w_os.append(offs(tws))
            b_os.append(offs(tbs))
            Gl.append(1)
            trl.append(0)
            dys_l.append(dzencs)
            actss.append(acts_m)
            wss_l.append(list(info["mixT"]))
            w_os.append(offs(info["mixW"]))
            b_os.append(offs(info["mixB"]))
            Gl.append(info["k"])
            trl.append(1)
            ext.mlp_narrow_bwd_multi(dys_l, actss, wss_l, arena2,
                                     w_os, b_os, Gl, trl)
            se_n = self._se_numel
            ext.reduce_arena(arena2, fg_c, S2, 0, se_n)
            ext.reduce_arena(arena_c, fg_c, S_c, se_n, -1)
        else:
            def cur_half(acts):
                return acts if dedup else [a[B:] for a in acts]
            if mlpctx is not None:
                self._mlp_bwd_arena(ext, dx0[:, :zc_dim].contiguous(),
                                    cur_half(se_saved["acts_c"]),
                                    cwsh,
                                    [w.grad for w in cws],
                                    [b.grad for b in cbs],
                                    fg_c, arena_c, S_c, ch_c)
            self._mlp_bwd_arena(ext, dlogits,
                                cur_half(se_saved["acts_t"]), twsh,
                                [w.grad for w in tws],
                                [b.grad for b in tbs],
                                fg_c, arena_c, S_c, ch_c)
            self._mlp_bwd_arena(ext, dzencs, acts_m, info["mixT"],
                                [w.grad for w in info["mixW"]],
                                [b.grad for b in info["mixB"]],
                                fg_c, arena_c, S_c, ch_c,
                                G=info["k"], transpose_w=1)
            ext.reduce_arena(arena_c, fg_c, S_c)
        if orig:
            # context-encoder backward: the only z_context gradient source
            # is the critic-loss concat half of dx0 (reference
            # retain_graph rule) — fused dx chain + grouped dwdb into the
            # context arena, Adam stepped at the end of the update
            cc = self._ctx_chain
            nlx = len(cc["ws16"])
            dzc = dx0[:, :zc_dim].contiguous()
            empty_h = states.new_empty(0, dtype=torch.bfloat16)
            youts_x = [acts_ctx[i + 1] for i in range(nlx - 1)] + [empty_h]
            dys_x = ext.mlp_chain_dx_bf16(
                dzc, list(cc["wt"]), youts_x, acts_ctx[0].shape[-1],
                [1] * (nlx - 1) + [0], 1, 1, -1, list(cc["dxp"]))
            fgx = self.context_group.flat_grad
            self.context_group.rebind_grads()
            basex = fgx.data_ptr()
            arena_x, S_x, ch_x = self._dw_arena(
                "ctx", self.context_group.numel, B)
            ext.dwdb_grouped_arena(
                list(dys_x), [acts_ctx[i] for i in range(nlx)], arena_x,
                [(l.weight.grad.data_ptr() - basex) // 4
                 for l in cc["lins"]],
                [(l.bias.grad.data_ptr() - basex) // 4
                 for l in cc["lins"]],
                1, S_x, ch_x)
            ext.reduce_arena(arena_x, fgx, S_x)
        self._dp_st = dict(
            chain=chain, states=states, sd=sd, zc16=zc16, sa=sa, lp=lp,
            ls_cat=ls_cat, B=B, eps=eps, tanh_u=tanh_u, lsr=lsr,
            acts_a=acts_a, ws_f32=ws_f32, bs_f32=bs_f32, la_det=la_det,
            T=T, use_w=use_w, closs=closs, orig=orig)

    @torch.no_grad()
    def _manual_seg2(self):
        from ..ops import native
        ext = native()
        st = self._dp_st
        chain = st["chain"]
        states, sd, zc16 = st["states"], st["sd"], st["zc16"]
        sa, lp, ls_cat, B = st["sa"], st["lp"], st["ls_cat"], st["B"]
        eps, tanh_u, lsr = st["eps"], st["tanh_u"], st["lsr"]
        acts_a, ws_f32, bs_f32 = st["acts_a"], st["ws_f32"], st["bs_f32"]
        la_det, T, use_w = st["la_det"], st["T"], st["use_w"]
        info = self._se_local
        nl_c = len(self._twin_local_bf16)
        nl_a = len(self._actor_ws_bf16)
        st["prolog"] = self._adam_prolog_all()
        # adam kernel refreshes the flat bf16 mirror
        self.critic_optimizer.step(pre_prologed=st["prolog"])
        self._refresh_mixT("critic")   # transposed mixture views still need it

        # ---- actor/alpha loss + manual backward -----------------------
        enc_c, _ = self._se_fwd_manual(info, states[:, :sd].to(torch.bfloat16),
                                       zc16)               # post-step SE
        if chain:
            # re-pack the POST-Adam critic (fwd + dx) in one launch
            ext.pack_weights_frag(
                list(self._twin_local_bf16) * 2,
                list(self._twin_local_fp) + list(self._twin_local_dxp),
                [2] * (2 * nl_c), [0] * nl_c + [1] * nl_c)
            ya, acts_f = self._chain_fwd(enc_c, sa, self._twin_local_bf16,
                                         self._twin_local[1], G=2,
                                         wps=self._twin_local_fp)
            aq1, aq2 = ya[0], ya[1]
        else:
            xa = torch.cat([enc_c, sa.to(torch.bfloat16)], dim=-1)
            aq1, aq2, acts_f = self._twin_fwd_manual(xa,
                                                     self._twin_local_bf16,
                                                     self._twin_local[1])
        # fwd kernel zeroes the alpha flat grad in passing (bwd atomics
        # target) — no separate fill launch
        al = ext.actor_alpha_loss_fwd(aq1, aq2, lp, ls_cat[B:], states,
                                      la_det, T, int(use_w), self.H_bar_f,
                                      self.alpha_group.flat_grad,
                                      self._aloss_ws)
        daq, dlp = ext.actor_alpha_loss_bwd2(
            aq1, aq2, lp, states, la_det, al, self.alpha_group.flat_grad,
            T, int(use_w), self.H_bar_f)
        if chain:
            empty_h = states.new_empty(0, dtype=torch.bfloat16)
            youts_f = [acts_f[i + 1] for i in range(nl_c - 1)] + [empty_h]
            outs = ext.mlp_chain_dx_bf16(
                daq, list(self._twin_local_wt), youts_f,
                acts_f[0].shape[-1], [1] * (nl_c - 1) + [0], 2, 0,
                enc_c.shape[1], list(self._twin_local_dxp))
            dx0a = outs[-1]
            dsa = dx0a              # twin heads summed inside squash bwd2
        else:
            dy = daq
            for i in range(nl_c - 1, 0, -1):
                act = 1 if i < nl_c - 1 else 0
                yout = acts_f[i + 1] if i < nl_c - 1 else acts_f[i]
                dy = ext.linear_bwd_dx_bf16(dy, self._twin_local_bf16[i],
                                            yout, act, 2, 0)
            dxa = ext.linear_bwd_dx_bf16(dy, self._twin_local_bf16[0],
                                         acts_f[1] if nl_c > 1 else acts_f[0],
                                         1 if nl_c > 1 else 0, 2, 1)
            dsa = dxa[:, enc_c.shape[1]:].float()
        dhead = ext.squashed_gaussian_bwd2(
            dsa, dlp, lsr[B:], ls_cat[B:], eps[B:], tanh_u[B:],
            float(self.actor.k))
        dy = dhead
        fg_a = self.actor_group.flat_grad
        arena_a, S_a, ch_a = self._dw_arena("actor",
                                            self.actor_group.numel, B)
        base_a = fg_a.data_ptr()
        if chain:
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
                [(w.grad.data_ptr() - base_a) // 4 for w in ws_f32],
                [(b.grad.data_ptr() - base_a) // 4 for b in bs_f32],
                1, S_a, ch_a)
        else:
            for i in range(nl_a - 1, -1, -1):
                act = 1 if i < nl_a - 1 else 0
                yout = (acts_a[i + 1][B:] if i < nl_a - 1
                        else acts_a[i][B:])
                ext.linear_bwd_dwdb_arena(
                    dy, acts_a[i][B:], yout, act, 1, arena_a,
                    (ws_f32[i].grad.data_ptr() - base_a) // 4,
                    (bs_f32[i].grad.data_ptr() - base_a) // 4, S_a, ch_a,
                    0)
                if i > 0:
                    dy = ext.linear_bwd_dx_bf16(dy, self._actor_ws_bf16[i],
                                                yout, act, 1, 1)
        ext.reduce_arena(arena_a, fg_a, S_a)
        st["al"] = al

    @torch.no_grad()
    def _manual_seg3(self):
        from ..ops.flat import FusedAdam as _FA
        st = self._dp_st
        _FA.step_many([self.actor_optimizer, self.log_alpha_optimizer],
                      rng_bump=(None if st.get("prolog")
                                else (self._rng_ctr if self._use_krng
                                      else None)),
                      pre_prologed=bool(st.get("prolog")))
        # self.alpha refreshed lazily outside the graph (see SACEngine)
        self._polyak_targets(mirror=self._target_bf16)
        self._refresh_mixT("target")
        if st["orig"]:
            # refreshes ctx mirror; prolog already ran in seg2
            self.context_encoder_optimizer.step(
                pre_prologed=bool(st.get("prolog")))
        self.tie_actor_state_encoder()
        closs, al = st["closs"], st["al"]
        return {"critic_loss": closs[6],  # summed in-kernel
                "actor_loss": al[0],
                "alpha_loss": al[2],
                "entropy": al[3]}

    @torch.no_grad()
    def publish_params(self) -> torch.Tensor:
        """[actor.state_encoder | actor head | trainable context params] —
        the CARE rollout policy needs the tied encoder and (original CARE)
        the trained context encoder (reference C3 publishes actor +
        context_encoder state_dicts, learner.py:412-422)."""
        parts = [self.actor_se_group.flat_data, self.actor_group.flat_data]
        if self.context_group is not None:
            parts.append(self.context_group.flat_data)
        return torch.cat(parts)

    # ------------------------------------------------------------------
    def checkpoint_state(self) -> Dict:
        def cpu_sd(m):
            return {k: v.cpu() for k, v in m.state_dict().items()}
        ctx_opt = (self.context_encoder_optimizer.state_dict()
                   if self.context_encoder_optimizer is not None else
                   {"state": {}, "param_groups": []})
        # MT10_Distributed_CARE/src/learner.py:178-198 key layout
        return {
            "update_iteration": self.update_iteration,
            "total_step": self.total_step,
            "context_encoder": cpu_sd(self.context_encoder),
            "context_encoder_optimizer": ctx_opt,
            "local_critic": cpu_sd(self.local_critic),
            "critic_optimizer": self.critic_optimizer.state_dict(),
            "target_critic": cpu_sd(self.target_critic),
            "actor": cpu_sd(self.actor),
            "actor_optimizer": self.actor_optimizer.state_dict(),
            "log_alpha": self.log_alpha.detach().cpu(),
            "log_alpha_optimizer": self.log_alpha_optimizer.state_dict(),
            "alpha": self.log_alpha.detach().exp().cpu(),
        }

    def load_checkpoint_state(self, ckpt: Dict) -> None:
        self.update_iteration = int(ckpt.get("update_iteration", 0))
        self.total_step = int(ckpt.get("total_step", 0))
        self.context_encoder.load_state_dict(ckpt["context_encoder"])
        self.local_critic.load_state_dict(ckpt["local_critic"])
        self.target_critic.load_state_dict(ckpt["target_critic"])
        self.actor.load_state_dict(ckpt["actor"])
        with torch.no_grad():
            self.log_alpha.copy_(ckpt["log_alpha"].to(self.device))
        self.alpha = self.log_alpha.exp().detach()
        for name, opt in (("critic_optimizer", self.critic_optimizer),
                          ("actor_optimizer", self.actor_optimizer),
                          ("log_alpha_optimizer", self.log_alpha_optimizer)):
            if name in ckpt:
                opt.load_state_dict(ckpt[name])
        if (self.context_encoder_optimizer is not None
                and "context_encoder_optimizer" in ckpt
                and ckpt["context_encoder_optimizer"].get("param_groups")):
            self.context_encoder_optimizer.load_state_dict(
                ckpt["context_encoder_optimizer"])
        self.refresh_bf16()  # bf16 mirrors (incl. mixT) track the masters
