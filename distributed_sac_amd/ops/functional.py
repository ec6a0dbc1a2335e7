"""Dispatching functional API for the SAC hot-path ops.

On CUDA(ROCm) tensors these route to the in-tree HIP/CDNA4 extension
(required — loud failure if missing); on CPU they compose the pure-torch
reference implementations from :mod:`.torch_ref`.

The custom autograd Functions wrap a whole fused region (an entire MLP, the
full squashed-Gaussian head) so the autograd graph has one node per region
instead of one per elementwise op — the backward chain is hand-written HIP.
"""

from __future__ import annotations

from typing import List, Optional, Sequence

import torch

from . import has_native, native, native_enabled
from . import torch_ref

ACT_NONE = 0
ACT_RELU = 1


def _use_native(t: torch.Tensor) -> bool:
    return t.is_cuda and native_enabled()


# ---------------------------------------------------------------------------
# Fused MLP (Linear+bias+ReLU chain, linear output layer)
# ---------------------------------------------------------------------------

class _FusedMLP(torch.autograd.Function):
    """Whole-MLP fused forward/backward on HIP.

    forward saves post-activation intermediates; ReLU backward masks come
    from the outputs themselves (relu(y)==0 <=> grad 0).
    """

    @staticmethod
    def forward(ctx, x: torch.Tensor, n_layers: int, grad_row_start: int,
                *wb):
        ws = wb[:n_layers]
        bs = wb[n_layers:]
        ext = native()
        acts: List[torch.Tensor] = [x]
        h = x
        for i in range(n_layers):
            act = ACT_RELU if i < n_layers - 1 else ACT_NONE
            h = ext.linear_act_fwd(h, ws[i], bs[i], act)
            acts.append(h)
        ctx.save_for_backward(*acts, *ws)
        ctx.n_layers = n_layers
        ctx.grad_row_start = grad_row_start
        return h

    @staticmethod
    def backward(ctx, grad_out: torch.Tensor):
        n = ctx.n_layers
        saved = ctx.saved_tensors
        acts = saved[: n + 1]
        ws = saved[n + 1:]
        ext = native()
        # rows below grad_row_start carry no gradient by construction (the
        # batched TD-side forward): slice them out of the backward chain
        # (row slices of row-major tensors stay contiguous)
        r0 = ctx.grad_row_start
        if r0:
            acts = [a[r0:] for a in acts]
            grad_out = grad_out[r0:]
        dy = grad_out.contiguous()
        dws: List[Optional[torch.Tensor]] = [None] * n
        dbs: List[Optional[torch.Tensor]] = [None] * n
        for i in range(n - 1, -1, -1):
            # mask==1 only for hidden layers (their saved act is post-ReLU)
            act = ACT_RELU if i < n - 1 else ACT_NONE
            if ctx.needs_input_grad[3 + i]:
                dw, db = ext.linear_bwd_dwdb(dy, acts[i], acts[i + 1], act)
                dws[i], dbs[i] = dw, db
            if i > 0:
                dy = ext.linear_bwd_dx(dy, ws[i], acts[i + 1], act)
        dx = ext.linear_bwd_dx(dy, ws[0], acts[1],
                               ACT_RELU if n > 1 else ACT_NONE) \
            if ctx.needs_input_grad[0] else None
        return (dx, None, None, *dws, *dbs)


def mlp_forward(x: torch.Tensor,
                weights: Sequence[torch.Tensor],
                biases: Sequence[torch.Tensor],
                grad_row_start: int = 0) -> torch.Tensor:
    """ReLU-hidden MLP with linear output (reference build_mlp semantics).

    grad_row_start>0: rows [0, start) are forward-only (their output grads
    are structurally zero — e.g. the TD half of a batched actor pass), so
    backward runs on the remaining rows only."""
    if _use_native(x):
        return _FusedMLP.apply(x, len(weights), grad_row_start,
                               *weights, *biases)
    return torch_ref.mlp_forward(x, weights, biases)


# ---------------------------------------------------------------------------
# Fused tanh-squashed Gaussian sample + log-prob
# ---------------------------------------------------------------------------

class _SquashedGaussian(torch.autograd.Function):
    @staticmethod
    def forward(ctx, mu, log_std_raw, eps, k: float):
        ext = native()
        action, log_prob, tanh_u, log_std = ext.squashed_gaussian_fwd(
            mu, log_std_raw, eps, float(k))
        ctx.save_for_backward(log_std_raw, log_std, eps, tanh_u)
        ctx.k = float(k)
        ctx.mark_non_differentiable(log_std)
        return action, log_prob, log_std

    @staticmethod
    def backward(ctx, grad_action, grad_log_prob, _grad_log_std):
        log_std_raw, log_std, eps, tanh_u = ctx.saved_tensors
        ext = native()
        dmu, dlog_std_raw = ext.squashed_gaussian_bwd(
            grad_action.contiguous(), grad_log_prob.contiguous(),
            log_std_raw, log_std, eps, tanh_u, ctx.k)
        return dmu, dlog_std_raw, None, None


def squashed_gaussian(mu: torch.Tensor, log_std_raw: torch.Tensor,
                      eps: torch.Tensor, k: float):
    """Returns (action, log_prob[B,1], clamped log_std).

    Gradients flow to mu and log_std_raw (pre-clamp), reproducing the
    reference's clamp-then-rsample autograd semantics
    (LunarLander…/src/model.py:45-59).
    """
    if _use_native(mu):
        return _SquashedGaussian.apply(mu, log_std_raw, eps, k)
    return torch_ref.squashed_gaussian(mu, log_std_raw, eps, k)


# ---------------------------------------------------------------------------
# Fused twin-MLP (grouped GEMMs: both Q networks in one launch per layer,
# stacked weights [2,N,K] — views into the critic's flat parameter buffer)
# ---------------------------------------------------------------------------

class _FusedTwinMLP(torch.autograd.Function):
    """Twin-Q MLP: q1,q2 from shared x through stacked weights [2,N,K].

    Layer 0 consumes the shared x (group-stride 0); later layers run on
    per-group activations [2,B,N].  Backward mirrors this: per-group dX for
    hidden layers, group-SUMMED dX at layer 0 (both Qs consume the same x —
    reference Critic.forward computes Q1(x), Q2(x) on one cat([s,a])).
    """

    @staticmethod
    def forward(ctx, x: torch.Tensor, n_layers: int, *wb):
        ws = wb[:n_layers]      # each [2, N, K]
        bs = wb[n_layers:]      # each [2, N]
        ext = native()
        acts = [x]
        h = x
        for i in range(n_layers):
            act = ACT_RELU if i < n_layers - 1 else ACT_NONE
            h = ext.linear_act_fwd_g(h, ws[i], bs[i], act, 2)
            acts.append(h)
        ctx.save_for_backward(*acts, *ws)
        ctx.n_layers = n_layers
        return h[0], h[1]

    @staticmethod
    def backward(ctx, dq1: torch.Tensor, dq2: torch.Tensor):
        n = ctx.n_layers
        saved = ctx.saved_tensors
        acts = saved[: n + 1]
        ws = saved[n + 1:]
        ext = native()
        dy = torch.stack([dq1, dq2], dim=0).contiguous()  # [2,B,Nout]
        dws = [None] * n
        dbs = [None] * n
        for i in range(n - 1, -1, -1):
            act = ACT_RELU if i < n - 1 else ACT_NONE
            if ctx.needs_input_grad[2 + i]:
                dw, db = ext.linear_bwd_dwdb_g(dy, acts[i], acts[i + 1],
                                               act, 2)
                dws[i], dbs[i] = dw, db
            if i > 0:
                dy = ext.linear_bwd_dx_g(dy, ws[i], acts[i + 1], act, 2, 0)
        dx = None
        if ctx.needs_input_grad[0]:
            dx = ext.linear_bwd_dx_g(dy, ws[0], acts[1],
                                     ACT_RELU if n > 1 else ACT_NONE, 2, 1)
        return (dx, None, *dws, *dbs)


def twin_mlp_forward(x: torch.Tensor, stacked_ws, stacked_bs):
    """Both Q heads in one grouped launch per layer.  stacked_ws[i] is a
    [2,N,K] view into the critic flat parameter buffer."""
    return _FusedTwinMLP.apply(x, len(stacked_ws), *stacked_ws, *stacked_bs)


# ---------------------------------------------------------------------------
# Fused SAC loss heads (reference math in kernel K6; one graph node per loss
# instead of ~15 torch glue kernels each).
# ---------------------------------------------------------------------------

class _CriticLoss(torch.autograd.Function):
    """(loss1, loss2) = (optionally task-weighted) MSE vs TD target.

    Matches reference Critic.cal_loss (MT10…MTSAC/src/model.py:157-196):
    w_i = softmax(-exp(log_alpha))[t_i] renormalized; loss = mean(w*l).
    """

    @staticmethod
    def forward(ctx, q1, q2, y, states, log_alpha_det, num_tasks: int,
                use_weighted: bool):
        ext = native()
        out = ext.critic_loss_fwd(q1.contiguous(), q2.contiguous(),
                                  y.contiguous(), states, log_alpha_det,
                                  num_tasks, int(use_weighted))[0]
        ctx.save_for_backward(q1, q2, y, states, log_alpha_det, out)
        ctx.meta = (num_tasks, use_weighted)
        return out[0], out[1]

    @staticmethod
    def backward(ctx, g1, g2):
        q1, q2, y, states, la, saved = ctx.saved_tensors
        T, use_w = ctx.meta
        gscale = torch.stack([g1.reshape(()), g2.reshape(())]).contiguous()
        dq1, dq2 = native().critic_loss_bwd(
            q1.contiguous(), q2.contiguous(), y.contiguous(), states, la,
            saved, gscale, T, int(use_w))
        return dq1, dq2, None, None, None, None, None


def critic_loss(q1, q2, y, states, log_alpha_det, num_tasks, use_weighted):
    if _use_native(q1):
        return _CriticLoss.apply(q1, q2, y, states, log_alpha_det,
                                 num_tasks, use_weighted)
    l1 = (y - q1) ** 2
    l2 = (y - q2) ** 2
    if use_weighted:
        w = torch_ref.task_weights(states[:, -num_tasks:],
                                   log_alpha_det.exp()).unsqueeze(-1)
        l1, l2 = w * l1, w * l2
    return l1.mean(), l2.mean()


class _ActorAlphaLoss(torch.autograd.Function):
    """(actor_loss, alpha_loss, entropy) fused.

    actor_loss = mean/weighted of -(min(q1,q2) - alpha_i * logp)
    alpha_loss = -mean(log_alpha[t_i] * (logp.detach() + H_bar))
    entropy    = mean(0.5*A*(1+log 2pi) + sum log_std)   [diagnostic]

    Gradients: actor_loss -> q1,q2,logp; alpha_loss -> log_alpha.
    (alpha is detached in the actor loss and logp in the alpha loss —
    reference learner.py:316, model.py:80-116.)
    """

    @staticmethod
    def forward(ctx, q1, q2, logp, log_stds, states, log_alpha,
                num_tasks: int, use_weighted: bool, H_bar: float):
        ext = native()
        out = ext.actor_alpha_loss_fwd(
            q1.contiguous(), q2.contiguous(), logp.contiguous(),
            log_stds.contiguous(), states, log_alpha.detach(), num_tasks,
            int(use_weighted), H_bar)
        ctx.save_for_backward(q1, q2, logp, states, log_alpha.detach(), out)
        ctx.meta = (num_tasks, use_weighted, H_bar)
        actor_l, alpha_l, entropy = out[0], out[2], out[3]
        ctx.mark_non_differentiable(entropy)
        return actor_l, alpha_l, entropy

    @staticmethod
    def backward(ctx, gp, gal, _gent):
        q1, q2, logp, states, la, saved = ctx.saved_tensors
        T, use_w, H_bar = ctx.meta
        gscale = torch.stack([gp.reshape(()), gal.reshape(())]).contiguous()
        daq1, daq2, dlp, dla = native().actor_alpha_loss_bwd(
            q1.contiguous(), q2.contiguous(), logp.contiguous(), states, la,
            saved, gscale, T, int(use_w), H_bar)
        return daq1, daq2, dlp, None, None, dla, None, None, None


def actor_alpha_loss(q1, q2, logp, log_stds, states, log_alpha, num_tasks,
                     use_weighted, H_bar):
    """Returns (actor_loss, alpha_loss, entropy)."""
    if _use_native(q1):
        return _ActorAlphaLoss.apply(q1, q2, logp, log_stds, states,
                                     log_alpha, num_tasks, use_weighted,
                                     float(H_bar))
    qmin = torch.min(q1, q2)
    if num_tasks > 1:
        one_hots = states[:, -num_tasks:]
        alpha = torch_ref.gather_log_alpha(one_hots,
                                           log_alpha).exp().detach()
        la_g = torch_ref.gather_log_alpha(one_hots, log_alpha)
    else:
        alpha = log_alpha.exp().detach()
        la_g = log_alpha
    pl = -(qmin - alpha * logp)
    if use_weighted:
        w = torch_ref.task_weights(states[:, -num_tasks:],
                                   log_alpha.exp().detach()).unsqueeze(-1)
        pl = w * pl
    actor_loss = pl.mean()
    alpha_loss = -(la_g * (logp.detach() + H_bar)).mean()
    entropy = torch_ref.entropy_from_log_std(log_stds)
    return actor_loss, alpha_loss, entropy


# ---------------------------------------------------------------------------
# Non-differentiable / glue ops
# ---------------------------------------------------------------------------

def td_target(rewards, dones, q1_t, q2_t, next_logp, alpha, gamma, reward_scale):
    if _use_native(rewards) and has_native():
        alpha_t = alpha if torch.is_tensor(alpha) else torch.full_like(rewards, float(alpha))
        if alpha_t.numel() == 1:
            alpha_t = alpha_t.expand_as(rewards).contiguous()
        return native().td_target(rewards, dones, q1_t, q2_t, next_logp,
                                  alpha_t, float(gamma), float(reward_scale))
    return torch_ref.td_target(rewards, dones, q1_t, q2_t, next_logp,
                               alpha, gamma, reward_scale)


def polyak_(target_params, source_params, tau: float) -> None:
    tps = list(target_params)
    sps = list(source_params)
    if tps and _use_native(tps[0]) and len(tps) == 1:
        native().polyak_(tps[0], sps[0], float(tau))
        return
    torch_ref.polyak_(tps, sps, tau)


# Pure-torch glue (tiny matmuls/reductions — not worth custom kernels; the
# heavy path is the fused MLP/squash/Adam/Polyak kernels):
task_weights = torch_ref.task_weights
gather_log_alpha = torch_ref.gather_log_alpha
entropy_from_log_std = torch_ref.entropy_from_log_std
batched_linear = torch_ref.batched_linear
attention_pool = torch_ref.attention_pool


# ---------------------------------------------------------------------------
# bf16 mixed-precision GEMM paths (fp32 master weights + bf16 compute
# mirrors; fp32 accumulation and fp32 weight grads).  The bf16 mirrors ride
# along as non-tracked args; the fp32 parameters are the autograd inputs so
# dW/db accumulate into the flat fp32 gradient buffers.
# ---------------------------------------------------------------------------

class _FusedMLPBF16(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x: torch.Tensor, n_layers: int, grad_row_start: int,
                ws_bf16, *wb_f32):
        ext = native()
        bs = wb_f32[n_layers:]
        xh = x.to(torch.bfloat16)
        acts = [xh]
        h = xh
        for i in range(n_layers):
            last = i == n_layers - 1
            act = ACT_NONE if last else ACT_RELU
            h = ext.linear_act_fwd_bf16(h, ws_bf16[i], bs[i].contiguous(),
                                        act, 1, 1 if last else 0)
            acts.append(h)
        ctx.save_for_backward(*acts[:-1])
        ctx.ws_bf16 = ws_bf16
        ctx.n_layers = n_layers
        ctx.grad_row_start = grad_row_start
        return h  # fp32

    @staticmethod
    def backward(ctx, grad_out: torch.Tensor):
        n = ctx.n_layers
        acts = list(ctx.saved_tensors)  # bf16: input + hidden outputs
        ws = ctx.ws_bf16
        ext = native()
        r0 = ctx.grad_row_start
        if r0:
            acts = [a[r0:] for a in acts]
            grad_out = grad_out[r0:]
        dy = grad_out.contiguous().to(torch.bfloat16)
        dws = [None] * n
        dbs = [None] * n
        for i in range(n - 1, -1, -1):
            act = ACT_RELU if i < n - 1 else ACT_NONE
            yout = acts[i + 1] if i < n - 1 else acts[i]  # dummy when act=0
            dw, db = ext.linear_bwd_dwdb_bf16(dy, acts[i], yout, act, 1)
            dws[i], dbs[i] = dw, db
            if i > 0:
                dy = ext.linear_bwd_dx_bf16(dy, ws[i], yout, act, 1, 1)
        dx = None
        if ctx.needs_input_grad[0]:
            dx = ext.linear_bwd_dx_bf16(
                dy, ws[0], acts[1] if n > 1 else acts[0],
                ACT_RELU if n > 1 else ACT_NONE, 1, 1).to(torch.float32)
        return (dx, None, None, None, *dws, *dbs)


def mlp_forward_bf16(x, ws_f32, bs_f32, ws_bf16, grad_row_start: int = 0):
    return _FusedMLPBF16.apply(x, len(ws_f32), grad_row_start,
                               tuple(ws_bf16), *ws_f32, *bs_f32)


class _FusedTwinMLPBF16(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x: torch.Tensor, n_layers: int, ws_bf16, *wb_f32):
        ext = native()
        bs = wb_f32[n_layers:]
        xh = x.to(torch.bfloat16)
        acts = [xh]
        h = xh
        for i in range(n_layers):
            last = i == n_layers - 1
            act = ACT_NONE if last else ACT_RELU
            h = ext.linear_act_fwd_bf16(h, ws_bf16[i], bs[i].contiguous(),
                                        act, 2, 1 if last else 0)
            acts.append(h)
        ctx.save_for_backward(*acts[:-1])
        ctx.ws_bf16 = ws_bf16
        ctx.n_layers = n_layers
        return h[0], h[1]  # fp32 [B, 1] each

    @staticmethod
    def backward(ctx, dq1, dq2):
        n = ctx.n_layers
        acts = list(ctx.saved_tensors)
        ws = ctx.ws_bf16
        ext = native()
        dy = torch.stack([dq1, dq2], dim=0).contiguous().to(torch.bfloat16)
        dws = [None] * n
        dbs = [None] * n
        for i in range(n - 1, -1, -1):
            act = ACT_RELU if i < n - 1 else ACT_NONE
            yout = acts[i + 1] if i < n - 1 else acts[i]
            # forward args: (x, n_layers, ws_bf16_tuple, *ws_f32, *bs_f32)
            # -> ws_f32[i] is input index 3+i (the non-tensor tuple at 2
            # always reports False — indexing 2+i silently skipped layer
            # 0's dW)
            if ctx.needs_input_grad[3 + i]:
                dw, db = ext.linear_bwd_dwdb_bf16(dy, acts[i], yout, act, 2)
                dws[i], dbs[i] = dw, db
            if i > 0:
                dy = ext.linear_bwd_dx_bf16(dy, ws[i], yout, act, 2, 0)
        dx = None
        if ctx.needs_input_grad[0]:
            dx = ext.linear_bwd_dx_bf16(
                dy, ws[0], acts[1] if n > 1 else acts[0],
                ACT_RELU if n > 1 else ACT_NONE, 2, 1).to(torch.float32)
        return (dx, None, None, *dws, *dbs)


def twin_mlp_forward_bf16(x, stacked_ws_f32, stacked_bs_f32, ws_bf16):
    return _FusedTwinMLPBF16.apply(x, len(ws_bf16), tuple(ws_bf16),
                                   *stacked_ws_f32, *stacked_bs_f32)


class _GroupedMLPBF16(torch.autograd.Function):
    """CARE mixture-of-k-encoders as grouped bf16 GEMMs (grid.z = k).

    Weights are the reference's batched-Linear parameters W (k,in,out) /
    b (k,1,out) (state_encoder.Linear); compute uses TRANSPOSED bf16
    mirrors (k,out,in) matching the GEMM kernel's [G,N,K] layout.  Layer 0
    consumes shared x; later layers per-group [k,B,*].  Output [k,B,out]
    fp32.  Backward runs full-batch (upstream grads are zero outside the
    grad-carrying rows; the mixture layers are 50-wide — cheaper than
    slicing per-group activations contiguous).
    """

    @staticmethod
    def forward(ctx, x: torch.Tensor, n_layers: int, k: int, ws_bf16T,
                *wb_f32):
        ext = native()
        bs = wb_f32[n_layers:]
        xh = x.to(torch.bfloat16)
        acts = [xh]
        h = xh
        for i in range(n_layers):
            last = i == n_layers - 1
            h = ext.linear_act_fwd_bf16(
                h, ws_bf16T[i], bs[i].view(k, -1).contiguous(),
                0 if last else 1, k, 1 if last else 0)
            acts.append(h)
        ctx.save_for_backward(*acts[:-1])
        ctx.ws_bf16T = ws_bf16T
        ctx.n_layers = n_layers
        ctx.k = k
        return h

    @staticmethod
    def backward(ctx, grad_out: torch.Tensor):
        n, k = ctx.n_layers, ctx.k
        acts = list(ctx.saved_tensors)
        ws = ctx.ws_bf16T
        ext = native()
        dy = grad_out.contiguous().to(torch.bfloat16)
        dws = [None] * n
        dbs = [None] * n
        for i in range(n - 1, -1, -1):
            act = ACT_RELU if i < n - 1 else ACT_NONE
            yout = acts[i + 1] if i < n - 1 else acts[i]
            # fwd args: (x, n_layers, k, ws_bf16T, *W, *b) -> W[i] at 4+i
            if ctx.needs_input_grad[4 + i]:
                dwT, db = ext.linear_bwd_dwdb_bf16(dy, acts[i], yout, act, k)
                dws[i] = dwT.permute(0, 2, 1)     # -> (k, in, out)
                dbs[i] = db.view(k, 1, -1)
            if i > 0:
                dy = ext.linear_bwd_dx_bf16(dy, ws[i], yout, act, k, 0)
        return (None, None, None, None, *dws, *dbs)


def grouped_mlp_bf16(x, W_f32_list, b_f32_list, ws_bf16T_list, k: int):
    return _GroupedMLPBF16.apply(x, len(W_f32_list), k,
                                 tuple(ws_bf16T_list), *W_f32_list,
                                 *b_f32_list)
