#!/usr/bin/env python3
"""Round-2 GPU training + quality-eval session on the full MI355X-native
stack: GPU-resident vectorised envs (batched env-step kernel), cached-models
fused SGD, reference-tuned PPO schedule — then evaluate the learned policy
against the SIX heuristic baselines over full episodes.

SiP-ML is evaluated at its configured static partition cap (the reference
actor's ``max_partitions_per_op`` parameter, ``sip_ml.py:4``) so it is
behaviourally distinct from MaxParallelism (with the cap left
at the env max the two are the same policy).

Env:  WORKLOAD=small_graphs|medium_graphs  ITERS=N  OUT=dir  EVAL_STEPS=N
"""
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch

from ddls_amd.envs import RampJobPartitioningEnvironment
from ddls_amd.envs.actors import ACTORS
from ddls_amd.models.gnn import GNNPolicy
from ddls_amd.rl.engine_env import EngineVectorEnv
from ddls_amd.rl.ppo import PPOConfig, PPOTrainer
from ddls_amd.runtime.checkpointer import Checkpointer
from ddls_amd.runtime.loops import EvalLoop, PolicyActor
from ddls_amd.utils import seed_everything

WORKLOAD = os.environ.get("WORKLOAD", "medium_graphs")
ITERS = int(os.environ.get("ITERS", "300"))
OUT = os.environ.get("OUT", f"gpurun_out/session_{WORKLOAD}")
EVAL_STEPS = int(os.environ.get("EVAL_STEPS", "0")) or 100000  # full episode
# per-workload exploration override (the reference's entropy_coeff was tuned
# on its own workload; medium_graphs needs longer exploration)
ENTROPY = float(os.environ.get("ENTROPY", "0.003"))
LR = float(os.environ.get("LR", "2.785e-4"))
DATA = os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
                    "data", WORKLOAD)


def make_env():
    return RampJobPartitioningEnvironment(
        reuse_jobs_generator=True,
        topology_config={"type": "ramp", "kwargs": {
            "num_communication_groups": 4,
            "num_racks_per_communication_group": 4,
            "num_servers_per_rack": 2,
            "num_channels": 1,
            "total_node_bandwidth": 1.6e12,
            "intra_gpu_propagation_latency": 50e-9,
            "worker_io_latency": 100e-9}},
        node_config={"type_1": {"num_nodes": 32, "workers_config": [
            {"num_workers": 1, "worker": "ddls_amd.devices.A100"}]}},
        jobs_config={"path_to_files": DATA,
                     "replication_factor": 1000,
                     "job_sampling_mode": "remove_and_repeat",
                     "job_interarrival_time_dist": {
                         "_target_": "ddls_amd.distributions.Fixed",
                         "val": 1000},
                     "max_acceptable_job_completion_time_frac_dist": {
                         "_target_": "ddls_amd.distributions.Uniform",
                         "min_val": 0.1, "max_val": 1, "decimals": 2},
                     "num_training_steps": 50},
        max_partitions_per_op=16,
        min_op_run_time_quantum=0.01,
        pad_obs_kwargs=None,
        max_simulation_run_time=1e6)


def main():
    os.makedirs(OUT, exist_ok=True)
    device = torch.device("cuda:0" if torch.cuda.is_available() else "cpu")

    torch.manual_seed(0)
    policy = GNNPolicy(num_actions=17)
    policy = policy.to(device)
    venv = EngineVectorEnv(make_env, num_envs=256, device=device,
                           base_seed=1, verbose=True)
    trainer = PPOTrainer(venv, policy,
                         PPOConfig(train_batch_size=4096,
                                   sgd_minibatch_size=128, num_sgd_iter=50,
                                   entropy_coeff=ENTROPY, lr=LR),
                         device=device)
    log = []
    t0 = time.time()
    for i in range(ITERS):
        st = trainer.train(num_steps=16)
        log.append({k: st.get(k) for k in
                    ("iteration", "mean_reward", "episode_reward_mean",
                     "blocking_rate_mean", "kl", "entropy")})
        if (i + 1) % 25 == 0:
            print(f"iter {i+1}: reward {st['mean_reward']:.1f} "
                  f"entropy {st['entropy']:.2f} "
                  f"({trainer.total_env_steps} steps, {time.time()-t0:.0f}s)",
                  flush=True)
    train_s = time.time() - t0
    print(f"trained {trainer.total_env_steps} env steps in {train_s:.0f}s "
          f"({trainer.total_env_steps/train_s:,.0f} env-steps/s sustained)")
    with open(f"{OUT}/train_log.json", "w") as f:
        json.dump(log, f)
    ck = Checkpointer(OUT)
    ck_path = ck.write(trainer.state_dict(), index=ITERS)
    print(f"checkpoint: {ck_path}")

    # ---- quality eval: learned policy vs the six baselines, full episodes
    rows = {}

    def ev(name, actor):
        seed_everything(1799)
        env = make_env()
        r = EvalLoop(actor, env, max_steps=EVAL_STEPS).run(seed=1799)
        es = env.cluster.episode_stats
        arrived = max(es["num_jobs_arrived"], 1)
        rows[name] = {
            "episode_return": r["episode_return"],
            "mean_jct": r["mean_job_completion_time"],
            "jct_speedup": r["mean_job_completion_time_speedup"],
            "blocking_rate": es["num_jobs_blocked"] / arrived,
            "num_jobs_arrived": es["num_jobs_arrived"],
            "num_actor_steps": r["num_actor_steps"],
        }
        print(f"{name}: {json.dumps(rows[name])}", flush=True)

    ev("learned_gnn", PolicyActor(trainer.policy, device=device))
    ev("random", ACTORS["random"]())
    ev("no_parallelism", ACTORS["no_parallelism"]())
    ev("min_parallelism", ACTORS["min_parallelism"]())
    ev("max_parallelism", ACTORS["max_parallelism"]())
    ev("sip_ml", ACTORS["sip_ml"](max_partitions_per_op=8))
    ev("acceptable_jct", ACTORS["acceptable_jct"]())
    with open(f"{OUT}/eval.json", "w") as f:
        json.dump(rows, f, indent=2)

    best_heur = max((v["episode_return"] for k, v in rows.items()
                     if k != "learned_gnn"))
    print(f"\nlearned return {rows['learned_gnn']['episode_return']:.0f} vs "
          f"best heuristic {best_heur:.0f} -> "
          f"{'DOMINATES' if rows['learned_gnn']['episode_return'] > best_heur else 'BEHIND'}")


if __name__ == "__main__":
    main()
