"""Checkpoint round-trip + reference schema tests."""

import os

import pytest
import torch

from distributed_sac_amd.algo import SACEngine
from distributed_sac_amd.checkpoint import (load_actor_for_eval,
                                            load_checkpoint,
                                            load_into_engine,
                                            save_checkpoint)
from tests.test_engine import make_batch, small_cfg


def test_schema_ll(tmp_path):
    engine = SACEngine(small_cfg("sac"), "cpu")
    p = save_checkpoint(engine, str(tmp_path), update_iteration=123)
    assert os.path.basename(p) == "checkpoint_123.tar"
    ckpt = load_checkpoint(p)
    # LunarLander…/src/learner.py:144-163 key layout
    for key in ("update_iteration", "total_step", "local_critic_1",
                "local_critic_2", "target_critic_1", "target_critic_2",
                "actor", "critic_optimizer", "actor_optimizer", "log_alpha",
                "log_alpha_optimizer", "alpha"):
        assert key in ckpt, key
    assert "local_critic" not in ckpt
    assert ckpt["episode_idx"] == 123  # LL counter key (learner.py:144-163)
    # actor state_dict keys match the reference LL actor module
    assert "layer_intermediate.0.weight" in ckpt["actor"]
    assert "mu_log_std_layer.weight" in ckpt["actor"]


def test_schema_mtsac(tmp_path):
    engine = SACEngine(small_cfg("mtsac"), "cpu")
    p = save_checkpoint(engine, str(tmp_path), update_iteration=7)
    ckpt = load_checkpoint(p)
    # MT10_Distributed_MTSAC/src/learner.py:157-174 key layout
    for key in ("local_critic", "target_critic", "actor", "log_alpha"):
        assert key in ckpt
    assert "local_critic_1" not in ckpt
    assert ckpt["log_alpha"].shape == (4,)
    assert "Q_function_1.0.weight" in ckpt["local_critic"]


def test_roundtrip_identical_behavior(tmp_path):
    torch.manual_seed(0)
    cfg = small_cfg("mtsac")
    e1 = SACEngine(cfg, "cpu")
    for i in range(3):
        e1.update(make_batch(cfg, seed=i))
    p = save_checkpoint(e1, str(tmp_path))

    e2 = SACEngine(cfg, "cpu")
    load_into_engine(e2, p)
    assert e2.update_iteration == e1.update_iteration
    for (n, p1), (_, p2) in zip(e1.actor.named_parameters(),
                                e2.actor.named_parameters()):
        assert torch.equal(p1, p2), n
    # optimizer state restored: identical further updates
    batch = make_batch(cfg, seed=99)
    eps = [torch.randn(cfg.batch_size, cfg.action_dim) for _ in range(2)]
    e1._eps_queue = [e.clone() for e in eps]
    e2._eps_queue = [e.clone() for e in eps]
    m1 = e1.update({k: v.clone() for k, v in batch.items()})
    m2 = e2.update({k: v.clone() for k, v in batch.items()})
    assert abs(m1["critic_loss"] - m2["critic_loss"]) < 1e-6
    for (n, p1), (_, p2) in zip(e1.actor.named_parameters(),
                                e2.actor.named_parameters()):
        assert torch.allclose(p1, p2, atol=1e-7), n


def test_optimizer_statedict_is_torch_adam_compatible(tmp_path):
    """Our FusedAdam state dicts load into torch.optim.Adam and back."""
    cfg = small_cfg("sac")
    engine = SACEngine(cfg, "cpu")
    for i in range(2):
        engine.update(make_batch(cfg, seed=i))
    sd = engine.actor_optimizer.state_dict()
    ref_adam = torch.optim.Adam(engine.actor.parameters(), lr=cfg.lr_actor)
    ref_adam.load_state_dict(sd)  # must not raise
    sd2 = ref_adam.state_dict()
    engine.actor_optimizer.load_state_dict(sd2)  # and back
    assert engine.actor_optimizer.step_count == 2


def test_player_side_eval_load(tmp_path):
    cfg = small_cfg("sac")
    engine = SACEngine(cfg, "cpu")
    p = save_checkpoint(engine, str(tmp_path), update_iteration=5)
    from distributed_sac_amd.models import LLActor
    actor = LLActor(cfg.state_dim, cfg.action_dim, cfg.actor_hidden_dim)
    it = load_actor_for_eval(actor, p)
    assert it == 5
    x = torch.randn(3, cfg.state_dim)
    assert torch.allclose(actor(x)[0], engine.actor(x)[0], atol=1e-6)


REF_LL_CKPT = ("/root/reference/saved_models/LunarLander_Distributed_SAC/"
               "checkpoint_165000.tar")


@pytest.mark.skipif(not os.path.exists(REF_LL_CKPT),
                    reason="reference checkpoint not present on this host")
def test_load_actual_reference_checkpoint():
    """The REAL reference artifact (trained by the reference repo on a GTX
    1080, torch 1.x) must load into our engine and policy unchanged."""
    import numpy as np
    from distributed_sac_amd.config import load_variant
    from distributed_sac_amd.models import LLActor
    from distributed_sac_amd.workers.player import evaluate_checkpoint
    from distributed_sac_amd.workers.trainer import default_env_fn

    cfg = load_variant("sac")
    engine = SACEngine(cfg, "cpu")
    load_into_engine(engine, REF_LL_CKPT)
    assert engine.update_iteration == 165000
    assert float(engine.log_alpha) < -2.0  # trained temperature
    # optimizer state restored into the fused Adam
    assert engine.critic_optimizer.step_count > 0

    actor = LLActor(cfg.state_dim, cfg.action_dim, cfg.actor_hidden_dim)
    it = load_actor_for_eval(actor, REF_LL_CKPT)
    assert it == 0 or it == 165000  # LL stores episode_idx, not update_it
    x = torch.randn(7, 8)
    mu_e, _ = engine.actor(x)
    mu_a, _ = actor(x)
    assert torch.allclose(mu_e, mu_a, atol=1e-7)
    # engine can continue training from the reference state
    from tests.test_engine import make_batch
    cfg_small = cfg
    batch = {
        "states": torch.randn(32, 8), "actions": torch.rand(32, 2) * 2 - 1,
        "rewards": torch.randn(32, 1), "next_states": torch.randn(32, 8),
        "dones": torch.zeros(32, 1),
    }
    m = engine.update(batch)
    assert np.isfinite(m["critic_loss"])
    # and the eval-mode player runs it on an env
    out = evaluate_checkpoint(cfg, REF_LL_CKPT, default_env_fn, episodes=2)
    assert np.isfinite(out["mean_reward"])



def test_shipped_trained_checkpoint_evaluates():
    """The trained synthetic-VSAC artifact under saved_models/ loads and
    achieves near-optimal deterministic reward (r11 learning run)."""
    import json

    from distributed_sac_amd.config import Decoder, SACConfig
    from distributed_sac_amd.workers.player import evaluate_checkpoint
    from distributed_sac_amd.workers.trainer import default_env_fn

    root = os.path.join(os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))), "saved_models", "MT1_VSAC_synthetic_tiny")
    cfg = SACConfig.from_dict(
        json.load(open(os.path.join(root, "cfg.json")), cls=Decoder), "vsac")
    out = evaluate_checkpoint(cfg, os.path.join(root, "checkpoint_553159.tar"),
                              default_env_fn, episodes=3, seed=123)
    assert out["update_iteration"] == 553159
    assert out["mean_reward"] > -5.0          # random policy is ~-70
    assert out["success_rate"] == 1.0


def test_shipped_mtsac_checkpoint_evaluates():
    """The trained MT10-MTSAC artifact (reference 3x400 architecture, MT
    .tar schema) evaluates with success 1.0 on an arbitrary task."""
    import json

    from distributed_sac_amd.config import Decoder, SACConfig
    from distributed_sac_amd.workers.player import evaluate_checkpoint
    from distributed_sac_amd.workers.trainer import default_env_fn

    root = os.path.join(os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))), "saved_models", "MT10_MTSAC_synthetic")
    cfg = SACConfig.from_dict(
        json.load(open(os.path.join(root, "cfg.json")), cls=Decoder), "mtsac")
    out = evaluate_checkpoint(cfg, os.path.join(root, "checkpoint_70718.tar"),
                              default_env_fn, task_idx=7, episodes=2,
                              seed=321)
    assert out["update_iteration"] == 70718
    assert out["mean_reward"] > -10.0
    assert out["success_rate"] == 1.0


def test_shipped_ll_checkpoint_evaluates():
    """The trained LunarLander-SAC artifact (reference LL .tar schema,
    update_delay-thinned counters) evaluates near-optimally."""
    import json

    from distributed_sac_amd.config import Decoder, SACConfig
    from distributed_sac_amd.workers.player import evaluate_checkpoint
    from distributed_sac_amd.workers.trainer import default_env_fn

    root = os.path.join(os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))), "saved_models", "LunarLander_synthetic")
    cfg = SACConfig.from_dict(
        json.load(open(os.path.join(root, "cfg.json")), cls=Decoder), "sac")
    out = evaluate_checkpoint(
        cfg, os.path.join(root, "checkpoint_292428.tar"),
        default_env_fn, episodes=2, seed=77)
    assert out["update_iteration"] == 292428
    assert out["mean_reward"] > -5.0   # random policy ~-70


def test_gpu_trained_checkpoint_evaluates():
    """The MI355X-trained MT10-MTSAC artifact (saved_models/mtsac_gpu,
    858k grad steps in 8 min through the async stack) loads in the
    reference schema and acts successfully on the synthetic suite."""
    import numpy as np
    from distributed_sac_amd.config import load_variant
    from distributed_sac_amd.workers.player import evaluate_checkpoint
    from distributed_sac_amd.workers.trainer import default_env_fn
    path = os.path.join(os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))), "saved_models", "mtsac_gpu",
        "checkpoint_5148606.tar")
    assert os.path.exists(path)
    cfg = load_variant("mtsac")
    out = evaluate_checkpoint(cfg, path, default_env_fn, task_idx=2,
                              episodes=5, seed=3)
    assert out["update_iteration"] == 5148606
    assert out["success_rate"] >= 0.8
    assert np.isfinite(out["mean_reward"])
