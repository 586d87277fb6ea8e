#!/usr/bin/env python3
"""Train the PAC-ML GNN partitioner with PPO.

Replaces the reference's ``scripts/train_rllib_from_config.py:37``.  Single
GPU: ``python scripts/train.py``; data-parallel over N GPUs (RCCL/xGMI):
``torchrun --nproc-per-node N --master-addr 127.0.0.1 scripts/train.py``.
Dotted config overrides: ``python scripts/train.py algo.lr=1e-4
num_epochs=50``.
"""
import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from ddls_amd.parallel import init_distributed_from_env
from ddls_amd.runtime.config import (build_env_from_config,
                                     build_trainer_from_config, load_config)
from ddls_amd.runtime.loops import EpochLoop, EvalLoop, Launcher, PolicyActor
from ddls_amd.utils import seed_everything


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--config", default=os.path.join(
        os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
        "configs", "train_config.yaml"))
    ap.add_argument("overrides", nargs="*", help="dotted overrides a.b=c")
    args = ap.parse_args()

    cfg = load_config(args.config, overrides=args.overrides)
    rank = init_distributed_from_env()
    use_cuda = torch.cuda.is_available()
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    device = torch.device(f"cuda:{local_rank}" if use_cuda else "cpu")

    seed_everything(cfg.get("seed", 0) + rank)
    trainer = build_trainer_from_config(cfg, device=device)

    eval_fn = None
    save_dir = None
    if rank == 0:
        save_dir = os.path.join(cfg["experiment"].get("path_to_save", "runs"),
                                cfg["experiment"].get("name", "train"))
        eval_cfg = cfg.get("eval_config", {})

        def eval_fn():
            env = build_env_from_config(cfg)
            actor = PolicyActor(trainer.policy, device=device)
            loop = EvalLoop(actor, env)
            return loop.run(seed=eval_cfg.get("eval_seed", 1799))

    launcher = Launcher(EpochLoop(trainer),
                        num_epochs=cfg.get("num_epochs"),
                        max_actor_steps=cfg.get("max_actor_steps"),
                        evaluation_interval=cfg.get("evaluation_interval", 10),
                        eval_fn=eval_fn,
                        path_to_save=save_dir,
                        use_sqlite_database=cfg.get("use_sqlite_database", False),
                        verbose=(rank == 0))
    launcher.run()


if __name__ == "__main__":
    main()
