#!/usr/bin/env python3
"""Round-1 GPU training session: train the PAC-ML policy with the full
MI355X-native stack (HIP fused fwd+bwd, batched lookahead memo precompute,
worker-side rollouts), then evaluate vs the six heuristic baselines."""
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch

from bench import build_env_fn
from ddls_amd.cluster.batched_lookahead import precompute_lookahead_memos
from ddls_amd.envs.actors import ACTORS
from ddls_amd.models.gnn import GNNPolicy
from ddls_amd.rl.ppo import PPOConfig, PPOTrainer
from ddls_amd.rl.subproc_env import SubprocVectorEnv
from ddls_amd.runtime.checkpointer import Checkpointer
from ddls_amd.runtime.loops import EvalLoop, PolicyActor
from ddls_amd.utils import seed_everything

ITERS = int(os.environ.get("ITERS", "400"))
OUT = os.environ.get("OUT", "gpurun_out/train_session")

os.makedirs(OUT, exist_ok=True)
device = torch.device("cuda:0" if torch.cuda.is_available() else "cpu")

env_fn = build_env_fn()
venv = SubprocVectorEnv(env_fn, num_envs=64, num_workers=64, base_seed=1)
scratch = env_fn(); scratch.reset(seed=0)
ml, mi = precompute_lookahead_memos(scratch, device=device)
del scratch
venv.preload_memos(ml, mi)
torch.manual_seed(0)
policy = GNNPolicy(num_actions=17)
trainer = PPOTrainer(venv, policy,
                     PPOConfig(train_batch_size=1024, sgd_minibatch_size=128,
                               num_sgd_iter=8),
                     device=device)
log = []
t0 = time.time()
for i in range(ITERS):
    st = trainer.train(num_steps=16)
    rec = {k: st.get(k) for k in ("iteration", "mean_reward",
                                  "episode_reward_mean", "blocking_rate_mean",
                                  "kl", "entropy")}
    log.append(rec)
    if (i + 1) % 20 == 0:
        print(f"iter {i+1}: reward {st['mean_reward']:.1f} "
              f"entropy {st['entropy']:.2f} "
              f"({trainer.total_env_steps} steps, {time.time()-t0:.0f}s)",
              flush=True)
venv.close()
with open(f"{OUT}/train_log.json", "w") as f:
    json.dump(log, f)
ck = Checkpointer(OUT)
ck.write(trainer.state_dict(), index=ITERS)
print(f"trained {trainer.total_env_steps} env steps in {time.time()-t0:.0f}s")

# evaluate vs baselines
rows = {}
def ev(name, actor):
    seed_everything(1799)
    env = env_fn()
    r = EvalLoop(actor, env, max_steps=600).run(seed=1799)
    es = env.cluster.episode_stats
    arrived = max(es["num_jobs_arrived"], 1)
    rows[name] = {"episode_return": r["episode_return"],
                  "mean_jct": r["mean_job_completion_time"],
                  "jct_speedup": r["mean_job_completion_time_speedup"],
                  "blocking_rate": es["num_jobs_blocked"] / arrived}
ev("learned_gnn", PolicyActor(trainer.policy, device=device))
for nm in ("random", "max_parallelism", "acceptable_jct"):
    ev(nm, ACTORS[nm]())
print(json.dumps(rows, indent=2))
with open(f"{OUT}/eval.json", "w") as f:
    json.dump(rows, f, indent=2)
