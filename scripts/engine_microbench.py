#!/usr/bin/env python3
"""Microbench the batched env-step kernel on the PAC-ML bench workload:
B vectorised envs, random mask-valid actions, steps/sec of the engine alone
(no policy).  Usage: python scripts/engine_microbench.py [B] [steps]
"""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import torch


def main():
    B = int(sys.argv[1]) if len(sys.argv) > 1 else 256
    steps = int(sys.argv[2]) if len(sys.argv) > 2 else 300

    sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
    from bench import build_env_fn
    from ddls_amd.cluster.vec_engine import (compile_engine_spec,
                                             drain_episode_schedule)
    from ddls_amd.cluster.gpu_engine import GpuEngine

    t0 = time.perf_counter()
    env = build_env_fn()()
    env.reset(seed=0)
    spec = compile_engine_spec(env, lookahead_device="cuda:0"
                               if torch.cuda.is_available() else None)
    t1 = time.perf_counter()
    print(f"[engine] spec compile: {t1 - t0:.2f}s "
          f"({len(spec.models)} models, {len(spec.mds)} md entries)")

    gen = env.cluster.jobs_generator
    scheds = [drain_episode_schedule(gen, spec, seed=1 + 1000 * b)
              for b in range(B)]
    t2 = time.perf_counter()
    print(f"[engine] schedule drain x{B}: {t2 - t1:.2f}s "
          f"(n={scheds[0].n} arrivals)")

    dev = torch.device("cuda:0")
    eng = GpuEngine(spec, B=B, device=dev)
    for b in range(B):
        eng.reset_env(b, scheds[b])
    t3 = time.perf_counter()
    print(f"[engine] state alloc + reset: {t3 - t2:.2f}s")

    rng = np.random.RandomState(0)
    ep_done = 0

    def random_actions():
        mask = eng.T["obs_mask"].cpu().numpy()
        acts = np.zeros(B, dtype=np.int64)
        for b in range(B):
            valid = np.flatnonzero(mask[b])
            acts[b] = valid[rng.randint(len(valid))]
        return torch.as_tensor(acts, device=dev)

    # warmup
    for _ in range(10):
        eng.step(random_actions())
    torch.cuda.synchronize()
    t4 = time.perf_counter()
    n_resets = 0
    for i in range(steps):
        eng.step(random_actions())
        done = eng.T["done"].cpu().numpy()
        for b in np.flatnonzero(done):
            ep_done += 1
            n_resets += 1
            sched = drain_episode_schedule(gen, spec,
                                           seed=77 + 1000 * b + n_resets)
            scheds[b] = sched
            eng.reset_env(int(b), sched)
    torch.cuda.synchronize()
    dt = time.perf_counter() - t4
    total = steps * B
    print(f"[engine] {steps} batched steps x {B} envs in {dt:.3f}s = "
          f"{total / dt:,.0f} env-steps/s ({dt / steps * 1e3:.2f} ms/step; "
          f"{ep_done} episode resets)")

    # pure-kernel rate (no host-side action sampling / done scan)
    acts = random_actions()
    torch.cuda.synchronize()
    t5 = time.perf_counter()
    for i in range(steps):
        eng.step(acts)
    torch.cuda.synchronize()
    dt = time.perf_counter() - t5
    print(f"[engine] kernel-only: {total / dt:,.0f} env-steps/s "
          f"({dt / steps * 1e3:.3f} ms/step)")


if __name__ == "__main__":
    main()
