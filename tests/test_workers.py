"""Async multi-process topology tests (CPU): players -> queue -> learner
-> snapshot -> players, with clean shutdown."""

import numpy as np
import pytest
import torch

from distributed_sac_amd.workers.orchestrator import (DistributedTrainer,
                                                      default_task_partition)
from distributed_sac_amd.workers.player import (apply_flat_params,
                                                build_actor,
                                                evaluate_checkpoint)
from tests.test_trainer import tiny_cfg


def test_task_partition():
    parts = default_task_partition(10, 4)
    assert sorted(t for p in parts for t in p) == list(range(10))
    assert len(parts) == 4
    parts1 = default_task_partition(1, 3)
    assert all(len(p) == 1 for p in parts1)


def test_apply_flat_params_roundtrip():
    cfg = tiny_cfg("mtsac")
    a1 = build_actor(cfg)
    a2 = build_actor(cfg)
    flat = torch.nn.utils.parameters_to_vector(a1.parameters())
    apply_flat_params(a2, flat)
    x = torch.randn(4, cfg.mtobs_dim)
    assert torch.allclose(a1(x)[0], a2(x)[0], atol=1e-7)


@pytest.mark.timeout(300)
def test_distributed_trainer_end_to_end_cpu():
    cfg = tiny_cfg("sac")
    cfg.start_memory_len = 128
    cfg.random_step = 64
    cfg.update_delay = 2
    dt = DistributedTrainer(cfg, device="cpu", num_players=2,
                            chunk_steps=32, seed=0, use_graph=False)
    stats = dt.run(max_grad_steps=20, max_seconds=120)
    assert stats.get("grad_steps", 0) >= 20
    # reference update_delay semantics: iteration counter thinned
    assert stats["iterations"] == stats["grad_steps"] * cfg.update_delay
    assert stats["ingested"] >= cfg.start_memory_len
    # snapshot advanced: players could pull fresh weights
    assert dt.snapshot.iteration() > 0
    assert all(not p.is_alive() for p in dt.players)


@pytest.mark.timeout(300)
def test_evaluate_checkpoint(tmp_path):
    from distributed_sac_amd.algo import SACEngine
    from distributed_sac_amd.checkpoint import save_checkpoint
    from distributed_sac_amd.workers.trainer import default_env_fn
    cfg = tiny_cfg("mtsac")
    engine = SACEngine(cfg, "cpu")
    p = save_checkpoint(engine, str(tmp_path), update_iteration=11)
    out = evaluate_checkpoint(cfg, p, default_env_fn, task_idx=1,
                              episodes=2, seed=0)
    assert out["update_iteration"] == 11
    assert out["episodes"] == 2
    assert np.isfinite(out["mean_reward"])
    assert 0.0 <= out["success_rate"] <= 1.0
