#!/usr/bin/env python3
"""IMPALA (V-trace) / PG training session on the GPU-resident env engine —
demonstrates the non-PPO algorithms end-to-end on the flagship stack.
Env: ALGO=impala|pg ITERS=N OUT=dir"""
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch

from bench import build_env_fn
from ddls_amd.models.gnn import GNNPolicy
from ddls_amd.rl.engine_env import EngineVectorEnv
from ddls_amd.rl.impala import ImpalaConfig, ImpalaTrainer
from ddls_amd.rl.pg import PGConfig, PGTrainer

ALGO = os.environ.get("ALGO", "impala")
ITERS = int(os.environ.get("ITERS", "40"))
OUT = os.environ.get("OUT", f"gpurun_out/{ALGO}_session")


def main():
    os.makedirs(OUT, exist_ok=True)
    dev = torch.device("cuda:0")
    torch.manual_seed(0)
    policy = GNNPolicy(num_actions=17).to(dev)
    venv = EngineVectorEnv(build_env_fn(), num_envs=256, device=dev,
                           base_seed=1)
    if ALGO == "pg":
        tr = PGTrainer(venv, policy, PGConfig(train_batch_size=4096),
                       device=dev)
    else:
        tr = ImpalaTrainer(venv, policy, ImpalaConfig(train_batch_size=4096),
                           device=dev)
    log = []
    t0 = time.time()
    for i in range(ITERS):
        st = tr.train(num_steps=16)
        log.append({k: st.get(k) for k in ("iteration", "mean_reward",
                                           "total_loss", "mean_rho",
                                           "entropy")})
        if (i + 1) % 10 == 0:
            print(f"iter {i+1}: reward {st['mean_reward']:.1f} "
                  f"loss {st['total_loss']:.1f} "
                  f"rho {st.get('mean_rho', 0):.3f} "
                  f"entropy {st.get('entropy', 0):.2f}", flush=True)
    dt = time.time() - t0
    print(f"{ALGO}: {tr.total_env_steps} env steps in {dt:.0f}s = "
          f"{tr.total_env_steps/dt:,.0f} steps/s")
    with open(f"{OUT}/train_log.json", "w") as f:
        json.dump(log, f)
    torch.save({"policy": policy.state_dict()}, f"{OUT}/policy.pt")

    # greedy full-episode eval on the REAL env vs the two strongest
    # heuristics, same protocol as scripts/train_eval_session.py
    from ddls_amd.envs.actors import ACTORS
    from ddls_amd.runtime.loops import EvalLoop, PolicyActor
    from ddls_amd.utils import seed_everything

    def run_eval(actor, steps=1000):
        seed_everything(1799)
        env = build_env_fn()()
        r = EvalLoop(actor, env, max_steps=steps).run(seed=1799)
        es = env.cluster.episode_stats
        arrived = max(es["num_jobs_arrived"], 1)
        return {
            "episode_return": r["episode_return"],
            "mean_jct": r["mean_job_completion_time"],
            "jct_speedup": r["mean_job_completion_time_speedup"],
            "blocking_rate": es["num_jobs_blocked"] / arrived,
            "num_actor_steps": r["num_actor_steps"],
        }

    evals = {f"learned_{ALGO}": run_eval(
        PolicyActor(policy, device=torch.device("cpu")))}
    for name, kw in (("sip_ml", {"max_partitions_per_op": 8}),
                     ("acceptable_jct", {})):
        evals[name] = run_eval(ACTORS[name](**kw))
    print(json.dumps(evals, indent=1))
    with open(f"{OUT}/eval.json", "w") as f:
        json.dump(evals, f, indent=1)


if __name__ == "__main__":
    main()
