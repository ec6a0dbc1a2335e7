"""CLI launcher — the reference main.py equivalent.

Train:
    python -m distributed_sac_amd.launch --variant mtsac --players 4 \
        --device cuda:0 --max-grad-steps 100000 --save-dir saved_models/mtsac
Eval (reference is_train=False branch):
    python -m distributed_sac_amd.launch --variant mtsac --eval \
        --checkpoint saved_models/mtsac/checkpoint_60000.tar --episodes 50

Unlike the reference (hand-edited is_train flag and hardcoded paths,
MT10_Distributed_MTSAC/src/main.py:15-31), everything is a flag.  For
data-parallel learners launch under torchrun with one rank per GPU:
    torchrun --nproc-per-node 8 --master-addr 127.0.0.1 \
        -m distributed_sac_amd.launch --variant mtsac ...
"""

from __future__ import annotations

import argparse
import json
import os

import torch


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--variant", default="mtsac",
                    choices=["sac", "vsac", "mtsac", "care", "mt1_care"])
    ap.add_argument("--cfg", default=None, help="cfg JSON path (default: shipped)")
    ap.add_argument("--device", default=None)
    ap.add_argument("--players", type=int, default=2)
    ap.add_argument("--max-grad-steps", type=int, default=None)
    ap.add_argument("--max-seconds", type=float, default=None)
    ap.add_argument("--save-dir", default=None)
    ap.add_argument("--save-period", type=int, default=0)
    ap.add_argument("--logdir", default=None)
    ap.add_argument("--seed", type=int, default=0)
    ap.add_argument("--no-graph", action="store_true")
    ap.add_argument("--precision", choices=["fp32", "bf16"], default=None)
    ap.add_argument("--eval-every", type=int, default=None,
                    help="player success-eval period in episodes "
                         "(default: 40 for Meta-World variants, 0 for LL)")
    ap.add_argument("--env", default=None,
                    help="env name for make_env (default: synthetic)")
    # eval mode
    ap.add_argument("--eval", action="store_true")
    ap.add_argument("--checkpoint", default=None)
    ap.add_argument("--episodes", type=int, default=50)
    ap.add_argument("--render", action="store_true")
    ap.add_argument("--task", type=int, default=0)
    args = ap.parse_args()

    from .config import SACConfig, load_variant
    from .utils import MetricLogger
    from .workers.trainer import default_env_fn

    if args.cfg:
        variant = None if args.variant == "auto" else \
            ("care" if args.variant == "mt1_care" else args.variant)
        cfg = SACConfig.from_file(args.cfg, variant)
    else:
        cfg = load_variant(args.variant)

    env_fn = default_env_fn
    if args.env:
        from .envs import make_env

        def env_fn(c, task_idx, seed):  # noqa: F811
            return make_env(args.env, seed)

    if args.eval:
        from .workers.player import evaluate_checkpoint
        assert args.checkpoint, "--eval needs --checkpoint"
        out = evaluate_checkpoint(cfg, args.checkpoint, env_fn,
                                  task_idx=args.task,
                                  episodes=args.episodes,
                                  render=args.render, seed=args.seed)
        print(json.dumps(out))
        return

    device = args.device or ("cuda:0" if torch.cuda.is_available() else "cpu")
    ddp = None
    if int(os.environ.get("WORLD_SIZE", "1")) > 1:
        from .parallel import DataParallelGroup
        local_rank = int(os.environ.get("LOCAL_RANK", "0"))
        device = f"cuda:{local_rank}" if torch.cuda.is_available() else "cpu"
        if torch.cuda.is_available():
            torch.cuda.set_device(device)
        ddp = DataParallelGroup()

    from .workers.orchestrator import DistributedTrainer
    logger = MetricLogger(args.logdir, stdout=True) if args.logdir else \
        MetricLogger(None, stdout=True)
    logger.write_hyperparameters(cfg.raw)
    # reference MT mains pass eval_episode_idx=40 in training; success
    # semantics exist for the Meta-World variants only
    eval_every = 40 if cfg.variant in ("vsac", "mtsac", "care") else 0
    if args.eval_every is None:
        args.eval_every = eval_every
    dt = DistributedTrainer(cfg, device=device, num_players=args.players,
                            env_fn=env_fn, logger=logger,
                            save_dir=args.save_dir,
                            save_period=args.save_period, seed=args.seed,
                            use_graph=not args.no_graph, ddp=ddp,
                            precision=args.precision,
                            eval_every_episodes=args.eval_every)
    stats = dt.run(max_grad_steps=args.max_grad_steps,
                   max_seconds=args.max_seconds)
    print(json.dumps(stats))


if __name__ == "__main__":
    main()
