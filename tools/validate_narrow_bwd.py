"""Round-2 validation for the staged fused narrow-chain backward
(k_bf16_mlp_narrow_bwd, gated behind DSAC_NARROW_BWD=1).

Run ON A GPU BOX:
    python tools/validate_narrow_bwd.py

Compares modified-CARE engine updates with the gate off vs on (same
init/batches/eps): flat parameter groups must track to ~1e-3 after a few
Adam steps (the fused path changes only partial-sum association), and
prints a quick bench A/B.  If this passes, flip the gate default in
algo/care.py and add the comparison as a @pytest.mark.gpu test.
"""

import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def build(tmp, gate):
    os.environ["DSAC_NARROW_BWD"] = gate
    from distributed_sac_amd.algo.care import CAREEngine
    from tests.test_care import care_cfg
    torch.manual_seed(0)
    cfg = care_cfg(tmp, modified=True)
    return CAREEngine(cfg, "cuda:0", precision="bf16"), cfg


def main():
    assert torch.cuda.is_available()
    import tempfile
    tmp = tempfile.mkdtemp()
    e_off, cfg = build(tmp, "0")
    e_on, _ = build(tmp, "1")
    e_on.load_checkpoint_state(e_off.checkpoint_state())
    from tests.test_care import care_batch
    B, A = cfg.batch_size, cfg.action_dim
    for step in range(3):
        batch = {k: v.cuda() for k, v in care_batch(cfg, seed=step).items()}
        eps = [torch.randn(B, A, device="cuda") for _ in range(2)]
        os.environ["DSAC_NARROW_BWD"] = "0"
        e_off._eps_queue = [t.clone() for t in eps]
        e_off.update({k: v.clone() for k, v in batch.items()})
        os.environ["DSAC_NARROW_BWD"] = "1"
        e_on._eps_queue = [t.clone() for t in eps]
        e_on.update(batch)
    torch.cuda.synchronize()
    ok = True
    for name, g1, g2 in (("critic", e_off.critic_group, e_on.critic_group),
                         ("actor", e_off.actor_group, e_on.actor_group),
                         ("alpha", e_off.alpha_group, e_on.alpha_group)):
        d = (g1.flat_data - g2.flat_data).abs().max().item()
        print(f"{name}: max param diff {d:.3e}")
        ok = ok and d < 3e-3
    print("AGREEMENT", "PASS" if ok else "FAIL")
    if not ok:
        sys.exit(1)

    # quick A/B (eager, no graph — relative comparison only)
    def rate(gate):
        os.environ["DSAC_NARROW_BWD"] = gate
        eng, c = build(tmp, gate)
        batch = {k: v.cuda() for k, v in care_batch(c, seed=9).items()}
        for _ in range(20):
            eng.update({k: v.clone() for k, v in batch.items()})
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(200):
            eng.update({k: v.clone() for k, v in batch.items()})
        torch.cuda.synchronize()
        return 200 / (time.perf_counter() - t0)

    r_off, r_on = rate("0"), rate("1")
    print(f"eager updates/s: off={r_off:.1f} on={r_on:.1f} "
          f"({(r_on / r_off - 1) * 100:+.1f}%)")


if __name__ == "__main__":
    main()
