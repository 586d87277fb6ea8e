#!/usr/bin/env python3
"""IMPALA (V-trace) training session on the GPU-resident env engine —
demonstrates the second algorithm end-to-end on the flagship stack.
Env: ITERS=N OUT=dir"""
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np
import torch

from bench import build_env_fn
from ddls_amd.models.gnn import GNNPolicy
from ddls_amd.rl.engine_env import EngineVectorEnv
from ddls_amd.rl.impala import ImpalaConfig, ImpalaTrainer

ITERS = int(os.environ.get("ITERS", "40"))
OUT = os.environ.get("OUT", "gpurun_out/impala_session")


def main():
    os.makedirs(OUT, exist_ok=True)
    dev = torch.device("cuda:0")
    torch.manual_seed(0)
    policy = GNNPolicy(num_actions=17).to(dev)
    venv = EngineVectorEnv(build_env_fn(), num_envs=256, device=dev,
                           base_seed=1)
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
                  f"loss {st['total_loss']:.1f} rho {st['mean_rho']:.3f} "
                  f"entropy {st['entropy']:.2f}", flush=True)
    dt = time.time() - t0
    print(f"IMPALA: {tr.total_env_steps} env steps in {dt:.0f}s = "
          f"{tr.total_env_steps/dt:,.0f} steps/s")
    with open(f"{OUT}/train_log.json", "w") as f:
        json.dump(log, f)


if __name__ == "__main__":
    main()
