"""HIP kernel numerics tests vs the pure-torch fp32 oracle (run on MI355X).

Every op is compared against ops.torch_ref at fp32 tolerances (the GEMM
kernels use the exact f32-input MFMA, so only summation-order differences
remain)."""

import pytest
import torch

pytestmark = pytest.mark.gpu

from distributed_sac_amd.ops import functional as Fops  # noqa: E402
from distributed_sac_amd.ops import torch_ref as R  # noqa: E402


def req_native():
    from distributed_sac_amd import ops
    assert ops.has_native(), "HIP extension must be built on the GPU box"
    return ops.native()


SHAPES = [
    (1280, 400, 49),   # MTSAC actor layer 1
    (1280, 400, 400),  # hidden
    (1280, 8, 400),    # actor head
    (1280, 1, 400),    # critic head
    (256, 256, 10),    # LL critic layer 1
    (100, 7, 3),       # ragged edges
    (64, 64, 64),
    (1, 400, 53),      # B=1 player inference
]


@pytest.mark.parametrize("M,N,K", SHAPES)
@pytest.mark.parametrize("act", [0, 1])
def test_linear_act_fwd(M, N, K, act):
    ext = req_native()
    torch.manual_seed(0)
    x = torch.randn(M, K, device="cuda")
    w = torch.randn(N, K, device="cuda") / K ** 0.5
    b = torch.randn(N, device="cuda")
    y = ext.linear_act_fwd(x, w, b, act)
    ref = torch.nn.functional.linear(x, w, b)
    if act == 1:
        ref = torch.relu(ref)
    assert torch.allclose(y, ref, atol=1e-4, rtol=1e-4), \
        f"max err {(y - ref).abs().max().item()}"


@pytest.mark.parametrize("M,N,K", [(1280, 400, 400), (1280, 400, 49),
                                   (100, 7, 3), (256, 1, 260)])
@pytest.mark.parametrize("act", [0, 1])
def test_linear_bwd(M, N, K, act):
    ext = req_native()
    torch.manual_seed(1)
    x = torch.randn(M, K, device="cuda")
    w = torch.randn(N, K, device="cuda") / K ** 0.5
    b = torch.randn(N, device="cuda")
    y = torch.nn.functional.linear(x, w, b)
    yact = torch.relu(y) if act == 1 else y
    dy = torch.randn(M, N, device="cuda")

    dy_m = dy * (yact > 0) if act == 1 else dy
    ref_dx = dy_m @ w
    ref_dw = dy_m.t() @ x
    ref_db = dy_m.sum(0)

    dx = ext.linear_bwd_dx(dy, w, yact, act)
    dw, db = ext.linear_bwd_dwdb(dy, x, yact, act)
    assert torch.allclose(dx, ref_dx, atol=1e-4, rtol=1e-4)
    assert torch.allclose(dw, ref_dw, atol=2e-3, rtol=1e-4)  # K-dim = M=1280
    assert torch.allclose(db, ref_db, atol=2e-3, rtol=1e-4)


def test_fused_mlp_autograd_matches_torch():
    torch.manual_seed(2)
    from distributed_sac_amd.models import build_mlp
    m = build_mlp(49, 8, [400, 400, 400]).cuda()
    x = torch.randn(1280, 49, device="cuda", requires_grad=True)
    y = m(x)
    loss = (y ** 2).mean()
    loss.backward()
    g_native = {n: p.grad.clone() for n, p in m.named_parameters()}
    gx_native = x.grad.clone()

    # torch eager reference with identical weights (deep-copied so grads
    # do not accumulate into the shared parameter objects)
    import copy
    import torch.nn as nn
    seq = nn.Sequential(*[copy.deepcopy(l) for l in m]).cuda()
    x2 = x.detach().clone().requires_grad_(True)
    y2 = seq(x2)
    assert torch.allclose(y, y2, atol=1e-4, rtol=1e-4)
    loss2 = (y2 ** 2).mean()
    loss2.backward()
    for n, p in seq.named_parameters():
        assert torch.allclose(g_native[n], p.grad, atol=1e-4, rtol=1e-3), n
    assert torch.allclose(gx_native, x2.grad, atol=1e-4, rtol=1e-3)


def test_squashed_gaussian_fwd_bwd():
    torch.manual_seed(3)
    B, A, k = 1280, 4, 1.0
    mu = torch.randn(B, A, device="cuda", requires_grad=True)
    lsr = (torch.randn(B, A, device="cuda") * 5).requires_grad_(True)
    eps = torch.randn(B, A, device="cuda")

    a, lp, ls = Fops.squashed_gaussian(mu, lsr, eps, k)
    loss = a.sum() + 2.0 * lp.sum()
    loss.backward()
    gmu, glsr = mu.grad.clone(), lsr.grad.clone()

    mu2 = mu.detach().clone().requires_grad_(True)
    lsr2 = lsr.detach().clone().requires_grad_(True)
    a2, lp2, ls2 = R.squashed_gaussian(mu2, lsr2, eps, k)
    assert torch.allclose(a, a2, atol=1e-5, rtol=1e-5)
    assert torch.allclose(lp, lp2, atol=1e-4, rtol=1e-4)
    assert torch.allclose(ls, ls2, atol=1e-6)
    (a2.sum() + 2.0 * lp2.sum()).backward()
    assert torch.allclose(gmu, mu2.grad, atol=1e-4, rtol=1e-4)
    assert torch.allclose(glsr, lsr2.grad, atol=1e-4, rtol=1e-4)


def test_td_target_kernel():
    torch.manual_seed(4)
    B = 1280
    r, lp = torch.randn(B, 1, device="cuda"), torch.randn(B, 1, device="cuda")
    d = (torch.rand(B, 1, device="cuda") < 0.3).float()
    q1, q2 = torch.randn(B, 1, device="cuda"), torch.randn(B, 1, device="cuda")
    alpha = torch.rand(B, 1, device="cuda")
    y = Fops.td_target(r, d, q1, q2, lp, alpha, 0.99, 1.5)
    ref = R.td_target(r, d, q1, q2, lp, alpha, 0.99, 1.5)
    assert torch.allclose(y, ref, atol=1e-5)


def test_adam_kernel_matches_torch_adam():
    torch.manual_seed(5)
    ext = req_native()
    n = 100001
    p = torch.randn(n, device="cuda")
    p_ref = p.clone()
    m = torch.zeros(n, device="cuda")
    v = torch.zeros(n, device="cuda")
    ref_p = torch.nn.Parameter(p_ref)
    opt = torch.optim.Adam([ref_p], lr=3e-4)
    for step in range(1, 4):
        g = torch.randn(n, device="cuda")
        ext.adam_step_(p, g, m, v, step, 3e-4, 0.9, 0.999, 1e-8)
        ref_p.grad = g.clone()
        opt.step()
    assert torch.allclose(p, ref_p.detach(), atol=1e-6, rtol=1e-5), \
        (p - ref_p.detach()).abs().max().item()


def test_polyak_kernel():
    ext = req_native()
    t = torch.randn(12345, device="cuda")
    s = torch.randn(12345, device="cuda")
    ref = 0.005 * s + 0.995 * t
    ext.polyak_(t, s, 0.005)
    assert torch.allclose(t, ref, atol=1e-6)


def test_engine_gpu_matches_cpu():
    """Full SAC update on GPU (HIP kernels) vs CPU (torch oracle), identical
    weights/batch/eps."""
    from distributed_sac_amd.algo import SACEngine
    from tests.test_engine import make_batch, small_cfg
    torch.manual_seed(0)
    cfg = small_cfg("mtsac")
    e_cpu = SACEngine(cfg, "cpu")
    e_gpu = SACEngine(cfg, "cuda:0")
    # copy weights cpu -> gpu
    e_gpu.actor.load_state_dict(e_cpu.actor.state_dict())
    e_gpu.local_critic.load_state_dict(e_cpu.local_critic.state_dict())
    e_gpu.hard_copy_targets()
    e_cpu.hard_copy_targets()
    for step in range(3):
        batch = make_batch(cfg, seed=step)
        eps = [torch.randn(cfg.batch_size, cfg.action_dim) for _ in range(2)]
        e_cpu._eps_queue = [e.clone() for e in eps]
        e_gpu._eps_queue = [e.clone() for e in eps]
        m_cpu = e_cpu.update({k: v.clone() for k, v in batch.items()})
        m_gpu = e_gpu.update({k: v.cuda() for k, v in batch.items()})
    assert abs(m_cpu["critic_loss"] - m_gpu["critic_loss"]) < 1e-3
    assert abs(m_cpu["actor_loss"] - m_gpu["actor_loss"]) < 1e-3
    for (n, pc), (_, pg) in zip(e_cpu.actor.named_parameters(),
                                e_gpu.actor.named_parameters()):
        assert torch.allclose(pc, pg.cpu(), atol=1e-4, rtol=1e-3), n


def test_native_required_on_gpu():
    """GPU path must not silently fall back to eager torch."""
    from distributed_sac_amd import ops
    assert ops.native_enabled()
    assert ops.has_native()
    import distributed_sac_amd.ops._hip_ops as ext
    assert ext.__file__.endswith(".so")
    assert "distributed_sac_amd" in ext.__file__


def test_trainer_end_to_end_gpu():
    from distributed_sac_amd.workers import Trainer
    from tests.test_trainer import tiny_cfg
    torch.manual_seed(0)
    cfg = tiny_cfg("mtsac")
    tr = Trainer(cfg, device="cuda:0", seed=0)
    metrics = tr.train(env_steps_per_iter=40, updates_per_iter=2, iterations=3)
    assert metrics and metrics["critic_loss"] == metrics["critic_loss"]


def test_graph_captured_update_matches_eager():
    """hipGraph-captured update must track the eager path: run two engines
    from identical state, one graphed and one eager, on the same replay
    content; losses must stay finite and parameters close after N steps."""
    from distributed_sac_amd.algo import SACEngine
    from distributed_sac_amd.replay import ShardedReplay
    from tests.test_engine import small_cfg
    torch.manual_seed(0)
    cfg = small_cfg("mtsac")
    dev = "cuda:0"
    engine = SACEngine(cfg, dev)
    replay = ShardedReplay(4000, cfg.num_tasks, cfg.mtobs_dim,
                           cfg.action_dim, device=dev)
    for t in range(cfg.num_tasks):
        n = 256
        st = torch.randn(n, cfg.mtobs_dim, device=dev)
        oh = torch.zeros(n, cfg.num_tasks, device=dev)
        oh[:, t] = 1
        st[:, -cfg.num_tasks:] = oh
        replay.shards[t].append(
            st, torch.rand(n, cfg.action_dim, device=dev) * 2 - 1,
            torch.randn(n, 1, device=dev), st.clone(),
            torch.zeros(n, 1, device=dev))

    engine.capture(replay, cfg.batch_size)
    it0 = engine.update_iteration
    step0 = engine.critic_optimizer.step_count
    for _ in range(5):
        m = engine.graphed_update()
    torch.cuda.synchronize()
    assert engine.update_iteration == it0 + 5
    # device-side Adam step counter advanced once per replay
    assert engine.critic_optimizer.step_count == step0 + 5
    for k, v in m.items():
        val = float(v)
        assert val == val, f"NaN metric {k}"
    for p in (engine.actor_group.flat_data, engine.critic_group.flat_data):
        assert torch.isfinite(p).all()


def test_grouped_twin_fwd_matches_single():
    ext = req_native()
    torch.manual_seed(10)
    B, N, K = 1280, 400, 53
    x = torch.randn(B, K, device="cuda")
    w = torch.randn(2, N, K, device="cuda") / K ** 0.5
    b = torch.randn(2, N, device="cuda")
    y = ext.linear_act_fwd_g(x, w, b, 1, 2)
    assert y.shape == (2, B, N)
    for g in range(2):
        ref = torch.relu(torch.nn.functional.linear(x, w[g], b[g]))
        assert torch.allclose(y[g], ref, atol=1e-4, rtol=1e-4)
    # per-group x
    x2 = torch.randn(2, B, N, device="cuda")
    w2 = torch.randn(2, 64, N, device="cuda") / N ** 0.5
    b2 = torch.randn(2, 64, device="cuda")
    y2 = ext.linear_act_fwd_g(x2, w2, b2, 0, 2)
    for g in range(2):
        ref = torch.nn.functional.linear(x2[g], w2[g], b2[g])
        assert torch.allclose(y2[g], ref, atol=1e-4, rtol=1e-4)


def test_twin_mlp_autograd():
    """Stacked twin-MLP fwd+bwd vs two independent torch MLPs."""
    torch.manual_seed(11)
    from distributed_sac_amd.ops.functional import twin_mlp_forward
    B, K, H = 512, 53, 128
    dims = [(H, K), (H, H), (1, H)]
    ws = [torch.randn(2, n, k, device="cuda").div_(k ** 0.5).requires_grad_(True)
          for n, k in dims]
    bs = [torch.randn(2, n, device="cuda").mul_(0.1).requires_grad_(True)
          for n, _ in dims]
    x = torch.randn(B, K, device="cuda", requires_grad=True)
    q1, q2 = twin_mlp_forward(x, ws, bs)
    loss = (q1 ** 2).mean() + (q2 * 3).mean()
    loss.backward()

    x2 = x.detach().clone().requires_grad_(True)
    ws2 = [w.detach().clone().requires_grad_(True) for w in ws]
    bs2 = [b.detach().clone().requires_grad_(True) for b in bs]
    outs = []
    for g in range(2):
        h = x2
        for i in range(3):
            h = torch.nn.functional.linear(h, ws2[i][g], bs2[i][g])
            if i < 2:
                h = torch.relu(h)
        outs.append(h)
    assert torch.allclose(q1, outs[0], atol=1e-4, rtol=1e-4)
    assert torch.allclose(q2, outs[1], atol=1e-4, rtol=1e-4)
    loss2 = (outs[0] ** 2).mean() + (outs[1] * 3).mean()
    loss2.backward()
    assert torch.allclose(x.grad, x2.grad, atol=1e-4, rtol=1e-3)
    for i in range(3):
        assert torch.allclose(ws[i].grad, ws2[i].grad, atol=1e-4, rtol=1e-3), i
        assert torch.allclose(bs[i].grad, bs2[i].grad, atol=1e-4, rtol=1e-3), i


@pytest.mark.parametrize("use_w,T", [(0, 1), (0, 10), (1, 10)])
def test_fused_critic_loss(use_w, T):
    torch.manual_seed(12)
    from distributed_sac_amd.ops.functional import critic_loss
    from distributed_sac_amd.ops import torch_ref as TR
    B, D = 1280, 39 + T
    states = torch.randn(B, D, device="cuda")
    idx = torch.randint(0, T, (B,), device="cuda")
    states[:, -T:] = torch.nn.functional.one_hot(idx, T).float()
    q1 = torch.randn(B, 1, device="cuda", requires_grad=True)
    q2 = torch.randn(B, 1, device="cuda", requires_grad=True)
    y = torch.randn(B, 1, device="cuda")
    la = torch.randn(T, device="cuda") * 0.3
    l1, l2 = critic_loss(q1, q2, y, states, la, T, bool(use_w))
    (l1 + l2).backward()

    q1r = q1.detach().clone().requires_grad_(True)
    q2r = q2.detach().clone().requires_grad_(True)
    rl1 = (y - q1r) ** 2
    rl2 = (y - q2r) ** 2
    if use_w:
        w = TR.task_weights(states[:, -T:], la.exp()).unsqueeze(-1)
        rl1, rl2 = w * rl1, w * rl2
    rl1, rl2 = rl1.mean(), rl2.mean()
    assert torch.allclose(l1, rl1, atol=1e-5, rtol=1e-4)
    assert torch.allclose(l2, rl2, atol=1e-5, rtol=1e-4)
    (rl1 + rl2).backward()
    assert torch.allclose(q1.grad, q1r.grad, atol=1e-6, rtol=1e-4)
    assert torch.allclose(q2.grad, q2r.grad, atol=1e-6, rtol=1e-4)


@pytest.mark.parametrize("use_w,T", [(0, 1), (1, 10)])
def test_fused_actor_alpha_loss(use_w, T):
    torch.manual_seed(13)
    from distributed_sac_amd.ops.functional import actor_alpha_loss
    from distributed_sac_amd.ops import torch_ref as TR
    B, A, D = 1280, 4, 39 + T
    H_bar = -float(A)
    states = torch.randn(B, D, device="cuda")
    idx = torch.randint(0, T, (B,), device="cuda")
    states[:, -T:] = torch.nn.functional.one_hot(idx, T).float()
    q1 = torch.randn(B, 1, device="cuda", requires_grad=True)
    q2 = torch.randn(B, 1, device="cuda", requires_grad=True)
    lp = torch.randn(B, 1, device="cuda", requires_grad=True)
    ls = torch.randn(B, A, device="cuda")
    la = (torch.randn(T, device="cuda") * 0.3).requires_grad_(True)

    al, all_, ent = actor_alpha_loss(q1, q2, lp, ls, states, la, T,
                                     bool(use_w), H_bar)
    (al + all_).backward()

    q1r = q1.detach().clone().requires_grad_(True)
    q2r = q2.detach().clone().requires_grad_(True)
    lpr = lp.detach().clone().requires_grad_(True)
    lar = la.detach().clone().requires_grad_(True)
    if T > 1:
        oh = states[:, -T:]
        alpha = (oh @ lar.unsqueeze(0).t()).exp().detach()
        la_g = oh @ lar.unsqueeze(0).t()
    else:
        alpha = lar.exp().detach()
        la_g = lar
    pl = -(torch.min(q1r, q2r) - alpha * lpr)
    if use_w:
        w = TR.task_weights(states[:, -T:], lar.exp().detach()).unsqueeze(-1)
        pl = w * pl
    rel = pl.mean()
    rela = -(la_g * (lpr.detach() + H_bar)).mean()
    rent = TR.entropy_from_log_std(ls)
    assert torch.allclose(al, rel, atol=1e-5, rtol=1e-4)
    assert torch.allclose(all_, rela, atol=1e-5, rtol=1e-4)
    assert torch.allclose(ent, rent, atol=1e-4, rtol=1e-4)
    (rel + rela).backward()
    assert torch.allclose(q1.grad, q1r.grad, atol=1e-6, rtol=1e-4)
    assert torch.allclose(q2.grad, q2r.grad, atol=1e-6, rtol=1e-4)
    assert torch.allclose(lp.grad, lpr.grad, atol=1e-6, rtol=1e-4)
    assert torch.allclose(la.grad, lar.grad, atol=1e-5, rtol=1e-4)


def test_fused_replay_sample():
    from distributed_sac_amd.replay import ShardedReplay
    torch.manual_seed(14)
    T, B = 10, 1280
    r = ShardedReplay(100000, T, 49, 4, device="cuda")
    for t in range(T):
        n = 500 + 37 * t
        st = torch.randn(n, 49, device="cuda")
        st[:, 39:] = 0.0
        st[:, 39 + t] = 1.0  # tag rows by task
        r.shards[t].append(st, torch.randn(n, 4, device="cuda"),
                           torch.full((n, 1), float(t), device="cuda"),
                           st.clone(), torch.zeros(n, 1, device="cuda"))
    out = r.sample(B, graph_safe=True)
    assert out["states"].shape == (B, 49)
    per = B // T
    for t in range(T):
        seg = out["rewards"][t * per:(t + 1) * per]
        assert (seg == float(t)).all(), f"shard {t} mis-sampled"
        oh = out["states"][t * per:(t + 1) * per, 39:]
        assert (oh.argmax(dim=1) == t).all()


def test_td_target_mt_kernel():
    ext = req_native()
    torch.manual_seed(15)
    B, T = 1280, 10
    states = torch.randn(B, 49, device="cuda")
    idx = torch.randint(0, T, (B,), device="cuda")
    states[:, -T:] = torch.nn.functional.one_hot(idx, T).float()
    r = torch.randn(B, 1, device="cuda")
    d = (torch.rand(B, 1, device="cuda") < 0.3).float()
    q1 = torch.randn(B, 1, device="cuda")
    q2 = torch.randn(B, 1, device="cuda")
    lp = torch.randn(B, 1, device="cuda")
    la = torch.randn(T, device="cuda") * 0.5
    y = ext.td_target_mt(r, d, q1, q2, lp, states, la, T, 0.99, 1.5)
    alpha = la.exp()[idx].unsqueeze(-1)
    ref = 1.5 * r + 0.99 * (1 - d) * (torch.min(q1, q2) - alpha * lp)
    assert torch.allclose(y, ref, atol=1e-5, rtol=1e-5)


# ---------------------------------------------------------------------------
# bf16 mixed-precision path
# ---------------------------------------------------------------------------

def test_bf16_fragment_layout_identity():
    """A=I with asymmetric B: catches any transposed/miswired fragment map
    (guide rule: symmetric B would silently pass a row<->col swap)."""
    ext = req_native()
    M = N = K = 64
    A = torch.eye(64, device="cuda")
    B = (torch.arange(64.0, device="cuda").view(64, 1) * 100
         + torch.arange(64.0, device="cuda").view(1, 64))  # B[i,j]=100i+j
    # y = A @ W^T with W = B --> y[i,j] = B[j,i]
    y = ext.linear_act_fwd_bf16(A.to(torch.bfloat16),
                                B.to(torch.bfloat16),
                                torch.zeros(64, device="cuda"), 0, 1, 1)
    ref = B.t()
    assert torch.allclose(y, ref, atol=16.0, rtol=1e-2), \
        (y - ref).abs().max().item()  # bf16 quantizes 100i+j to ~1% rel


@pytest.mark.parametrize("M,N,K", [(1280, 400, 400), (1280, 400, 49),
                                   (2560, 16, 400), (100, 7, 3)])
def test_bf16_fwd(M, N, K):
    ext = req_native()
    torch.manual_seed(20)
    x = torch.randn(M, K, device="cuda")
    w = torch.randn(N, K, device="cuda") / K ** 0.5
    b = torch.randn(N, device="cuda")
    xh, wh = x.to(torch.bfloat16), w.to(torch.bfloat16)
    y = ext.linear_act_fwd_bf16(xh, wh, b, 1, 1, 1)  # fp32 out
    ref = torch.relu(torch.nn.functional.linear(
        xh.float(), wh.float(), b))
    assert torch.allclose(y, ref, atol=5e-2, rtol=2e-2), \
        (y - ref).abs().max().item()
    yh = ext.linear_act_fwd_bf16(xh, wh, b, 1, 1, 0)  # bf16 out
    assert torch.allclose(yh.float(), ref, atol=2e-1, rtol=2e-2)


def test_bf16_grouped_and_bwd():
    ext = req_native()
    torch.manual_seed(21)
    B, N, K, G = 1280, 400, 53, 2
    x = torch.randn(B, K, device="cuda").to(torch.bfloat16)
    w = (torch.randn(G, N, K, device="cuda") / K ** 0.5).to(torch.bfloat16)
    b = torch.randn(G, N, device="cuda")
    y = ext.linear_act_fwd_bf16(x, w, b, 1, G, 0)
    assert y.shape == (G, B, N) and y.dtype == torch.bfloat16
    for g in range(G):
        ref = torch.relu(torch.nn.functional.linear(x.float(), w[g].float(),
                                                    b[g]))
        assert torch.allclose(y[g].float(), ref, atol=2e-1, rtol=2e-2)

    dy = torch.randn(G, B, N, device="cuda").to(torch.bfloat16)
    # dx summed over groups, relu-masked by y
    dx = ext.linear_bwd_dx_bf16(dy, w, y, 1, G, 1)
    ref_dx = sum((dy[g].float() * (y[g].float() > 0)) @ w[g].float()
                 for g in range(G))
    assert torch.allclose(dx.float(), ref_dx, atol=1.5, rtol=3e-2), \
        (dx.float() - ref_dx).abs().max().item()
    # per-group dx
    dxp = ext.linear_bwd_dx_bf16(dy, w, y, 1, G, 0)
    for g in range(G):
        ref_g = (dy[g].float() * (y[g].float() > 0)) @ w[g].float()
        assert torch.allclose(dxp[g].float(), ref_g, atol=1.0, rtol=3e-2)
    # dwdb fp32 out
    dw, db = ext.linear_bwd_dwdb_bf16(dy, x, y, 1, G)
    assert dw.dtype == torch.float32
    for g in range(G):
        dy_m = dy[g].float() * (y[g].float() > 0)
        ref_dw = dy_m.t() @ x.float()
        ref_db = dy_m.sum(0)
        assert torch.allclose(dw[g], ref_dw, atol=2.0, rtol=3e-2), \
            (dw[g] - ref_dw).abs().max().item()
        assert torch.allclose(db[g], ref_db, atol=1.0, rtol=3e-2)


def test_bf16_engine_update_and_graph():
    """bf16 engine update: finite, tracks the fp32 engine loosely, graph-
    capturable, and the critic learns on a fixed batch."""
    from distributed_sac_amd.algo import SACEngine
    from distributed_sac_amd.replay import ShardedReplay
    from tests.test_engine import make_batch, small_cfg
    torch.manual_seed(0)
    cfg = small_cfg("mtsac")
    e32 = SACEngine(cfg, "cuda:0")
    e16 = SACEngine(cfg, "cuda:0", precision="bf16")
    e16.actor.load_state_dict(e32.actor.state_dict())
    e16.local_critic.load_state_dict(e32.local_critic.state_dict())
    e16.hard_copy_targets()
    e16.refresh_bf16()
    e32.hard_copy_targets()
    for step in range(3):
        batch = {k: v.cuda() for k, v in make_batch(cfg, seed=step).items()}
        eps = [torch.randn(cfg.batch_size, cfg.action_dim, device="cuda")
               for _ in range(2)]
        e32._eps_queue = [e.clone() for e in eps]
        e16._eps_queue = [e.clone() for e in eps]
        m32 = e32.update({k: v.clone() for k, v in batch.items()})
        m16 = e16.update({k: v.clone() for k, v in batch.items()})
    assert abs(m32["critic_loss"] - m16["critic_loss"]) < 0.25 + \
        0.25 * abs(m32["critic_loss"])
    for v in m16.values():
        assert v == v

    # graph capture with bf16 kernels
    replay = ShardedReplay(4000, cfg.num_tasks, cfg.mtobs_dim,
                           cfg.action_dim, device="cuda:0")
    for t in range(cfg.num_tasks):
        n = 256
        st = torch.randn(n, cfg.mtobs_dim, device="cuda")
        oh = torch.zeros(n, cfg.num_tasks, device="cuda")
        oh[:, t] = 1
        st[:, -cfg.num_tasks:] = oh
        replay.shards[t].append(
            st, torch.rand(n, cfg.action_dim, device="cuda") * 2 - 1,
            torch.randn(n, 1, device="cuda"), st.clone(),
            torch.zeros(n, 1, device="cuda"))
    e16.capture(replay, cfg.batch_size)
    losses = []
    for _ in range(30):
        m = e16.graphed_update()
        losses.append(float(m["critic_loss"]))
    assert all(v == v for v in losses)
    assert losses[-1] < losses[0] * 1.5  # not diverging


def test_manual_backward_matches_autograd_bf16():
    """The hand-rolled bf16 backward must track the autograd bf16 path
    (same kernels, same order — only the gradient bookkeeping differs)."""
    import os
    from distributed_sac_amd.algo import SACEngine
    from tests.test_engine import make_batch, small_cfg
    torch.manual_seed(0)
    cfg = small_cfg("mtsac")
    e_man = SACEngine(cfg, "cuda:0", precision="bf16")
    e_aut = SACEngine(cfg, "cuda:0", precision="bf16")
    e_aut.actor.load_state_dict(e_man.actor.state_dict())
    e_aut.local_critic.load_state_dict(e_man.local_critic.state_dict())
    e_aut.hard_copy_targets()
    e_aut.refresh_bf16()
    e_man.hard_copy_targets()
    e_man.refresh_bf16()
    os.environ["DSAC_NO_MANUAL"] = "1"
    first = None
    try:
        for step in range(3):
            batch = {k: v.cuda() for k, v in
                     make_batch(cfg, seed=step).items()}
            eps = [torch.randn(cfg.batch_size, cfg.action_dim,
                               device="cuda") for _ in range(2)]
            e_aut._eps_queue = [e.clone() for e in eps]
            m_aut = e_aut.update({k: v.clone() for k, v in batch.items()})
            os.environ["DSAC_NO_MANUAL"] = "0"
            e_man._eps_queue = [e.clone() for e in eps]
            m_man = e_man.update({k: v.clone() for k, v in batch.items()})
            os.environ["DSAC_NO_MANUAL"] = "1"
            if first is None:
                first = (m_aut, m_man)
    finally:
        os.environ.pop("DSAC_NO_MANUAL", None)
    # step 1: identical inputs, losses computed before any divergence can
    # feed back — tight (both paths still have in-path fp32 atomics only
    # in the alpha grad, which is applied AFTER these losses)
    assert abs(first[0]["critic_loss"] - first[1]["critic_loss"]) < 1e-4
    assert abs(first[0]["actor_loss"] - first[1]["actor_loss"]) < 1e-4
    # step 3: allow the atomic-ordering drift (alpha grads feed back)
    assert abs(m_aut["critic_loss"] - m_man["critic_loss"]) < 1e-2 + \
        2e-2 * abs(m_aut["critic_loss"])
    assert abs(m_aut["actor_loss"] - m_man["actor_loss"]) < 1e-2
    for (n, pa), (_, pm) in zip(e_aut.actor.named_parameters(),
                                e_man.actor.named_parameters()):
        assert torch.allclose(pa, pm, atol=1e-3, rtol=2e-2), n
    for (n, pa), (_, pm) in zip(e_aut.local_critic.named_parameters(),
                                e_man.local_critic.named_parameters()):
        assert torch.allclose(pa, pm, atol=1e-3, rtol=2e-2), n
    assert torch.allclose(e_aut.log_alpha, e_man.log_alpha, atol=1e-3)


def test_narrow_bwd_gate_equivalence(tmp_path):
    """The fused narrow-chain SE backward (default since round 2) must
    produce the same parameters as the per-layer arena path — only the
    partial-sum association differs (measured exact on MI355X)."""
    import os
    from distributed_sac_amd.algo import CAREEngine
    from tests.test_care import care_batch, care_cfg

    def make(gate):
        os.environ["DSAC_NARROW_BWD"] = gate
        torch.manual_seed(0)
        cfg = care_cfg(tmp_path, modified=True)
        return CAREEngine(cfg, "cuda:0", precision="bf16"), cfg

    try:
        e_off, cfg = make("0")
        e_on, _ = make("1")
        e_on.load_checkpoint_state(e_off.checkpoint_state())
        e_on.hard_copy_targets()
        e_off.hard_copy_targets()
        for step in range(3):
            batch = {k: v.cuda() for k, v in
                     care_batch(cfg, seed=step).items()}
            eps = [torch.randn(cfg.batch_size, cfg.action_dim,
                               device="cuda") for _ in range(2)]
            os.environ["DSAC_NARROW_BWD"] = "0"
            e_off._eps_queue = [e.clone() for e in eps]
            e_off.update({k: v.clone() for k, v in batch.items()})
            os.environ["DSAC_NARROW_BWD"] = "1"
            e_on._eps_queue = [e.clone() for e in eps]
            e_on.update({k: v.clone() for k, v in batch.items()})
    finally:
        os.environ.pop("DSAC_NARROW_BWD", None)
    for (n, pa), (_, pb) in zip(e_off.local_critic.named_parameters(),
                                e_on.local_critic.named_parameters()):
        assert torch.allclose(pa, pb, atol=1e-5), n
    for (n, pa), (_, pb) in zip(e_off.actor.named_parameters(),
                                e_on.actor.named_parameters()):
        assert torch.allclose(pa, pb, atol=1e-5), n
    assert torch.allclose(e_off.log_alpha, e_on.log_alpha, atol=1e-6)


def test_dp_segmented_graphs_world1():
    """Single-GPU validation of the data-parallel launch path: a forced
    world-1 RCCL group (AVG all-reduce == identity), segmented capture
    (3 hipGraphs with eager collectives between), 30 replays stable."""
    import torch.distributed as dist
    from distributed_sac_amd.algo import SACEngine
    from distributed_sac_amd.parallel import DataParallelGroup
    from distributed_sac_amd.replay import ShardedReplay
    from tests.test_engine import small_cfg

    cfg = small_cfg("mtsac")
    ddp = DataParallelGroup(device=torch.device("cuda:0"), force=True)
    assert ddp.enabled and ddp.world_size == 1
    try:
        # world-1 AVG all-reduce must be an identity (replica semantics)
        t = torch.randn(1000, device="cuda")
        t0 = t.clone()
        ddp.allreduce_grad_(t)
        assert torch.equal(t, t0)

        torch.manual_seed(0)
        engine = SACEngine(cfg, "cuda:0", precision="bf16")
        engine.attach_ddp(ddp)
        replay = ShardedReplay(4096, cfg.num_tasks, cfg.mtobs_dim,
                               cfg.action_dim, device="cuda:0", seed=3)
        for tsk in range(cfg.num_tasks):
            n = 256
            s = torch.randn(n, cfg.mtobs_dim, device="cuda")
            s[:, -cfg.num_tasks:] = 0
            s[:, -cfg.num_tasks + tsk] = 1
            replay.shards[tsk].append(
                s, torch.rand(n, cfg.action_dim, device="cuda") * 2 - 1,
                torch.randn(n, 1, device="cuda"), s.clone(),
                torch.zeros(n, 1, device="cuda"))
        engine.capture_dp(replay, cfg.batch_size)
        losses = []
        for _ in range(30):
            m = engine.dp_graphed_update()
            losses.append(float(m["critic_loss"]))
        assert all(v == v for v in losses)          # no NaN
        assert losses[-1] < losses[0] * 1.5          # not diverging
    finally:
        if dist.is_initialized():
            dist.destroy_process_group()


def test_chain_fwd_bitwise_vs_per_layer():
    """k_bf16_chain_fwd must be BITWISE-equal to the per-layer
    cat + cast + k_bf16_fwd decomposition (same MFMA tiling/order)."""
    from distributed_sac_amd import ops
    ext = ops.native()
    torch.manual_seed(0)
    M, dims, G = 1280, [53, 400, 400, 1], 2
    x1 = torch.randn(M, dims[0] - 4, device="cuda")
    x2 = torch.randn(M, 4, device="cuda")
    ws, bs, K = [], [], dims[0]
    for N in dims[1:]:
        ws.append((torch.randn(G, N, K, device="cuda") / K ** 0.5)
                  .to(torch.bfloat16).contiguous())
        bs.append(torch.randn(G, N, device="cuda"))
        K = N
    xh = torch.cat([x1, x2], dim=-1).to(torch.bfloat16)
    acts_ref, h = [xh], xh
    for i, w in enumerate(ws):
        last = i == len(ws) - 1
        h = ext.linear_act_fwd_bf16(h, w, bs[i], 0 if last else 1, G,
                                    1 if last else 0)
        acts_ref.append(h)
    out = ext.mlp_chain_fwd_bf16(x1, x2, ws, bs, 0, G, 1, 0, 0, 1)
    assert torch.equal(out[1], acts_ref[0])
    assert torch.equal(out[0], acts_ref[-1])
    for a, r in zip(out[2:], acts_ref[1:-1]):
        assert torch.equal(a, r)


def test_chain_dx_and_grouped_dwdb_vs_per_layer():
    """Fused dx chain + grouped dwdb vs the per-layer kernels."""
    from distributed_sac_amd import ops
    ext = ops.native()
    torch.manual_seed(1)
    G, M = 2, 512
    dims = [104, 400, 400, 1]
    ws, wts, youts, flags, K = [], [], [], [], dims[0]
    for i, N in enumerate(dims[1:]):
        last = i == len(dims) - 2
        w = (torch.randn(G, N, K, device="cuda") / 8).to(torch.bfloat16)
        w = w.contiguous()
        ws.append(w)
        wt = torch.empty(G, K, N, device="cuda", dtype=torch.bfloat16)
        ext.transpose_weights_bf16([w], [wt], [G])
        assert torch.equal(wt, w.transpose(1, 2).contiguous())
        wts.append(wt)
        flags.append(0 if last else 1)
        youts.append(torch.empty(0, device="cuda", dtype=torch.bfloat16)
                     if last else
                     (torch.randn(G, M, N, device="cuda").relu())
                     .to(torch.bfloat16).contiguous())
        K = N
    dy_last = (torch.randn(G, M, dims[-1], device="cuda") / 4) \
        .to(torch.bfloat16).contiguous()
    out = ext.mlp_chain_dx_bf16(dy_last, wts, youts, dims[0], flags, G,
                                1, 100)
    dys, dx0 = out[:-1], out[-1]
    # per-layer reference for masked dys via torch fp32
    dyp = dy_last.float()
    for l in range(len(ws) - 1, -1, -1):
        if flags[l]:
            dyp = dyp * (youts[l].float() != 0)
        ref = dyp.to(torch.bfloat16)
        d = (dys[l].float() - ref.float()).abs().max().item()
        assert d <= 2e-2, f"layer {l} masked dy diff {d}"
        dyp = torch.einsum("gmn,gnk->gmk", ref.float(), ws[l].float())
    ref_dx0 = dyp[..., 100:]
    assert (dx0 - ref_dx0).abs().max().item() <= \
        1e-2 * ref_dx0.abs().max().item() + 1e-3
    # grouped dwdb == per-layer arena path (bitwise)
    S, chunk = 8, (M + 7) // 8
    chunk = (chunk + 63) // 64 * 64
    layers = [(dims[i + 1], dims[i]) for i in range(len(dims) - 1)]
    xs = [(torch.randn(G, M, Kl, device="cuda") / 4).to(torch.bfloat16)
          .contiguous() for _, Kl in layers]
    numel = sum(G * (N * Kl + N) for N, Kl in layers)
    w_offs, b_offs, off = [], [], 0
    for N, Kl in layers:
        w_offs.append(off)
        off += G * N * Kl
        b_offs.append(off)
        off += G * N
    a1 = torch.zeros(S, numel, device="cuda")
    a2 = torch.zeros(S, numel, device="cuda")
    o1 = torch.zeros(numel, device="cuda")
    o2 = torch.zeros(numel, device="cuda")
    dys_c = [d.contiguous() for d in dys]
    ext.dwdb_grouped_arena(dys_c, xs, a1, w_offs, b_offs, G, S, chunk)
    ext.reduce_arena(a1, o1, S, 0, -1)
    for i in range(len(layers)):
        ext.linear_bwd_dwdb_arena(dys_c[i], xs[i], dys_c[i], 0, G, a2,
                                  w_offs[i], b_offs[i], S, chunk, 0)
    ext.reduce_arena(a2, o2, S, 0, -1)
    assert torch.equal(o1, o2)


@pytest.mark.gpu
def test_chunked_capture_matches_sequential_replays():
    """A graph capturing chunk=2 updates must produce the SAME parameter
    trajectory as two chunk=1 replays: the device-side RNG counter and
    Adam/bias-correction state advance inside the graph, so given the
    same torch seed both engines consume identical replay indices and
    eps draws."""
    from distributed_sac_amd.algo import SACEngine
    from distributed_sac_amd.replay import ShardedReplay
    from tests.test_engine import small_cfg

    def build(seed, chunk):
        torch.manual_seed(seed)
        cfg = small_cfg("mtsac")
        eng = SACEngine(cfg, "cuda:0", precision="bf16")
        replay = ShardedReplay(4000, cfg.num_tasks, cfg.mtobs_dim,
                               cfg.action_dim, device="cuda:0")
        g = torch.Generator(device="cuda").manual_seed(7)
        for t in range(cfg.num_tasks):
            n = 256
            st = torch.randn(n, cfg.mtobs_dim, device="cuda", generator=g)
            oh = torch.zeros(n, cfg.num_tasks, device="cuda")
            oh[:, t] = 1
            st[:, -cfg.num_tasks:] = oh
            replay.shards[t].append(
                st,
                torch.rand(n, cfg.action_dim, device="cuda",
                           generator=g) * 2 - 1,
                torch.randn(n, 1, device="cuda", generator=g), st.clone(),
                torch.zeros(n, 1, device="cuda"))
        eng.capture(replay, cfg.batch_size, warmup_iters=1, chunk=chunk)
        return eng

    e1 = build(123, chunk=1)
    if not e1._use_krng:
        pytest.skip("counter RNG disabled (DSAC_KRNG=0)")
    e2 = build(123, chunk=2)
    # Replay-to-replay trajectories are NOT bitwise reproducible: the
    # alpha-grad atomicAdd order varies per execution and early-step Adam
    # amplifies ~1e-6 gradient noise into ~1e-3 parameter steps (sign
    # flips against bias-corrected step_size).  A control of TWO
    # IDENTICAL chunk=1 engines diverges by 0.9e-3/2.4e-3 (actor, 2/8
    # updates) — the chunked engine must stay inside that same envelope.
    for i in range(4):
        e1.graphed_update()
        e1.graphed_update()
        e2.graphed_update()
        torch.cuda.synchronize()
        da = float((e1.actor_group.flat_data
                    - e2.actor_group.flat_data).abs().max())
        dc = float((e1.critic_group.flat_data
                    - e2.critic_group.flat_data).abs().max())
        assert da < 2e-2 and dc < 2e-2, (i, da, dc)
    # loss values are far more parameter-sensitive than the params
    # themselves this early in training — finite + positive is the
    # meaningful check; the parameter envelope above is the invariant
    for m in (e1._graph_metrics, e2._graph_metrics):
        v = float(m["critic_loss"])
        assert v == v and 0.0 < v < 100.0
    assert e1.update_iteration == 8 and e2.update_iteration == 8


@pytest.mark.gpu
def test_counter_rng_reproducible_and_distinct():
    """Counter RNG: same seed -> identical trajectories; consecutive
    updates draw DIFFERENT noise (the counter advances in-graph); and
    the squash eps written back is standard-normal-ish."""
    from distributed_sac_amd import ops
    ext = ops.native()
    dev = "cuda:0"
    B, A = 4096, 4
    mu = torch.zeros(B, A, device=dev)
    lsr = torch.zeros(B, A, device=dev)
    ctr = torch.tensor([42], dtype=torch.int64, device=dev)
    eps1 = torch.empty(B, A, device=dev)
    ext.squashed_gaussian_fwd(mu, lsr, eps1, 1.0, ctr)
    eps_same = torch.empty(B, A, device=dev)
    ext.squashed_gaussian_fwd(mu, lsr, eps_same, 1.0, ctr)
    assert torch.equal(eps1, eps_same)          # same counter -> same draw
    ctr += 1
    eps2 = torch.empty(B, A, device=dev)
    ext.squashed_gaussian_fwd(mu, lsr, eps2, 1.0, ctr)
    assert not torch.equal(eps1, eps2)          # bumped counter -> fresh
    for e in (eps1, eps2):
        assert abs(float(e.mean())) < 0.05
        assert abs(float(e.std()) - 1.0) < 0.05
        assert float(e.abs().max()) < 6.0
    # replay_sample counter path: in-range rows, deterministic
    T, cap, Ds, Da = 2, 64, 8, 3
    g = torch.Generator(device=dev).manual_seed(0)
    f_s = torch.randn(T, cap, Ds, device=dev, generator=g)
    f_a = torch.randn(T, cap, Da, device=dev, generator=g)
    f_r = torch.randn(T, cap, 1, device=dev, generator=g)
    f_d = torch.zeros(T, cap, 1, device=dev)
    sizes = torch.full((T,), float(cap), device=dev)
    empty = torch.empty(0, device=dev)
    o1 = ext.replay_sample(f_s, f_a, f_r, f_s, f_d, sizes, empty, 32, ctr)
    o2 = ext.replay_sample(f_s, f_a, f_r, f_s, f_d, sizes, empty, 32, ctr)
    assert torch.equal(o1[0], o2[0])
    # every sampled row must exist in its task's storage
    for i in range(32):
        t = min(i // 16, T - 1)
        row = o1[0][i]
        d = (f_s[t] - row).abs().sum(dim=1)
        assert float(d.min()) < 1e-6
