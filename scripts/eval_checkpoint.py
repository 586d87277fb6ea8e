#!/usr/bin/env python3
"""Evaluate a trained policy checkpoint (reference
``scripts/test_rllib_from_config.py:44``)."""
import argparse
import json
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from ddls_amd.runtime.config import build_env_from_config, load_config
from ddls_amd.runtime.loops import EvalLoop, PolicyActor
from ddls_amd.utils import seed_everything


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("checkpoint", help="path to checkpoint-N file")
    ap.add_argument("--config", default=os.path.join(
        os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
        "configs", "train_config.yaml"))
    ap.add_argument("--episodes", type=int, default=3)
    ap.add_argument("--stochastic", action="store_true")
    ap.add_argument("overrides", nargs="*")
    args = ap.parse_args()

    cfg = load_config(args.config, overrides=args.overrides)
    eval_seed = cfg.get("eval_config", {}).get("eval_seed", 1799)
    actor = PolicyActor.from_checkpoint(
        args.checkpoint,
        num_actions=cfg.get("env_config", {}).get("max_partitions_per_op", 16) + 1,
        config=cfg.get("model", {}).get("custom_model_config"),
        deterministic=not args.stochastic)

    all_results = []
    for ep in range(args.episodes):
        seed_everything(eval_seed + ep)
        env = build_env_from_config(cfg)
        loop = EvalLoop(actor, env)
        r = loop.run(seed=eval_seed + ep)
        result = {
            "episode": ep,
            "blocking_rate": r["blocking_rate"],
            "acceptance_rate": r["acceptance_rate"],
            "mean_job_completion_time": r["mean_job_completion_time"],
            "mean_job_completion_time_speedup":
                r["mean_job_completion_time_speedup"],
            "episode_return": r["episode_return"],
        }
        all_results.append(result)
        print(json.dumps(result), flush=True)


if __name__ == "__main__":
    main()
