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
    def forward(ctx, x: torch.Tensor, n_layers: int, *wb):
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
        return h

    @staticmethod
    def backward(ctx, grad_out: torch.Tensor):
        n = ctx.n_layers
        saved = ctx.saved_tensors
        acts = saved[: n + 1]
        ws = saved[n + 1:]
        ext = native()
        dy = grad_out.contiguous()
        dws: List[Optional[torch.Tensor]] = [None] * n
        dbs: List[Optional[torch.Tensor]] = [None] * n
        for i in range(n - 1, -1, -1):
            # mask==1 only for hidden layers (their saved act is post-ReLU)
            act = ACT_RELU if i < n - 1 else ACT_NONE
            dw, db = ext.linear_bwd_dwdb(dy, acts[i], acts[i + 1], act)
            dws[i], dbs[i] = dw, db
            if i > 0:
                dy = ext.linear_bwd_dx(dy, ws[i], acts[i + 1], act)
        dx = ext.linear_bwd_dx(dy, ws[0], acts[1],
                               ACT_RELU if n > 1 else ACT_NONE) \
            if ctx.needs_input_grad[0] else None
        return (dx, None, *dws, *dbs)


def mlp_forward(x: torch.Tensor,
                weights: Sequence[torch.Tensor],
                biases: Sequence[torch.Tensor]) -> torch.Tensor:
    """ReLU-hidden MLP with linear output (reference build_mlp semantics)."""
    if _use_native(x):
        return _FusedMLP.apply(x, len(weights), *weights, *biases)
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
