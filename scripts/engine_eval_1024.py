#!/usr/bin/env python3
"""BASELINE configs[3]+[4]: the 1024-worker RAMP topology on the GPU-resident
env engine — batched env-step throughput, and a blocking-rate eval under a
Poisson (exponential-interarrival) multi-job queue with the heuristic
pipeline, all on-device.

Usage: python scripts/engine_eval_1024.py [--workers 1024] [--envs 64]
"""
import argparse
import json
import math
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np
import torch

from ddls_amd.cluster.gpu_engine import GpuEngine
from ddls_amd.cluster.vec_engine import (compile_engine_spec,
                                         drain_episode_schedule)
from ddls_amd.envs import RampJobPartitioningEnvironment


def make_env(workers: int, interarrival_mean: float, max_sim: float):
    if workers == 1024:
        shape = (16, 16, 4)
    elif workers == 32:
        shape = (4, 4, 2)
    else:
        raise ValueError(workers)
    return RampJobPartitioningEnvironment(
        reuse_jobs_generator=True,
        topology_config={"type": "ramp", "kwargs": {
            "num_communication_groups": shape[0],
            "num_racks_per_communication_group": shape[1],
            "num_servers_per_rack": shape[2],
            "num_channels": 1,
            "total_node_bandwidth": 1.6e12,
            "intra_gpu_propagation_latency": 50e-9,
            "worker_io_latency": 100e-9}},
        node_config={"type_1": {"num_nodes": workers, "workers_config": [
            {"num_workers": 1, "worker": "ddls_amd.devices.A100"}]}},
        jobs_config={"path_to_files": os.path.join(
            os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
            "data", "small_graphs"),
            "replication_factor": 1000,
            "job_sampling_mode": "remove_and_repeat",
            # configs[4]: Poisson arrival process
            "job_interarrival_time_dist": {
                "_target_": "ddls_amd.distributions.Exponential",
                "mean": interarrival_mean},
            "max_acceptable_job_completion_time_frac_dist": {
                "_target_": "ddls_amd.distributions.Uniform",
                "min_val": 0.1, "max_val": 1, "decimals": 2},
            "num_training_steps": 50},
        max_partitions_per_op=16,
        min_op_run_time_quantum=0.01,
        pad_obs_kwargs=None,
        max_simulation_run_time=max_sim)


def heuristic_actions(name, eng, scheds, rng):
    """Vectorised heuristic actors straight off the engine's device mask."""
    mask = eng.T["obs_mask"].cpu().numpy()
    sched_idx = eng.T["obs_sched"].cpu().numpy()
    B, A = mask.shape
    acts = np.zeros(B, dtype=np.int64)
    for b in range(B):
        valid = np.flatnonzero(mask[b])
        if len(valid) <= 1:
            acts[b] = valid[0] if len(valid) else 0
            continue
        if name == "random":
            acts[b] = int(rng.choice(valid[1:]))
        elif name == "max_parallelism":
            acts[b] = int(valid[-1])
        elif name == "acceptable_jct":
            k = int(sched_idx[b])
            sched = scheds[b]
            seq = eng.spec.models[int(sched.model_id[k])].seq_total
            acc = float(sched.max_acceptable[k])
            want = int(math.ceil(seq / acc)) if acc > 0 else int(valid[-1])
            choice = int(valid[-1])
            for a in valid:
                if a >= want:
                    choice = int(a)
                    break
            acts[b] = choice
        else:
            raise ValueError(name)
    return torch.as_tensor(acts, device=eng.T["obs_mask"].device)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--workers", type=int, default=1024)
    ap.add_argument("--envs", type=int, default=64)
    ap.add_argument("--interarrival", type=float, default=60.0)
    ap.add_argument("--max-sim", type=float, default=30000.0)
    ap.add_argument("--bench-steps", type=int, default=200)
    args = ap.parse_args()

    dev = torch.device("cuda:0")
    t0 = time.perf_counter()
    env = make_env(args.workers, args.interarrival, args.max_sim)
    env.reset(seed=0)
    spec = compile_engine_spec(env, lookahead_device=dev)
    gen = env.cluster.jobs_generator
    print(f"[1024] spec compile {time.perf_counter()-t0:.1f}s "
          f"(W={spec.W}, {len(spec.mds)} md entries)")

    results = {}
    # Poisson arrival counts vary per episode: size the job log for the tail
    jobs_cap = int(args.max_sim / args.interarrival * 1.5) + 64
    for actor in ("acceptable_jct", "max_parallelism", "random"):
        scheds = [drain_episode_schedule(gen, spec, seed=100 + b)
                  for b in range(args.envs)]
        eng = GpuEngine(spec, B=args.envs, device=dev, n_jobs_cap=jobs_cap)
        for b in range(args.envs):
            eng.reset_env(b, scheds[b])
        rng = np.random.RandomState(0)
        done_stats = []
        t1 = time.perf_counter()
        steps = 0
        while len(done_stats) < args.envs and steps < 20000:
            acts = heuristic_actions(actor, eng, scheds, rng)
            eng.step(acts)
            steps += 1
            done = eng.T["done"].cpu().numpy()
            for b in np.flatnonzero(done):
                done_stats.append(eng.episode_stats(int(b)))
                eng.T["status"][int(b)] = 0   # leave done (no reset)
        dt = time.perf_counter() - t1
        blocking = float(np.mean([s["blocking_rate"] for s in done_stats]))
        jcts = [np.mean(s["job_completion_time"]) for s in done_stats
                if s["num_jobs_completed"]]
        results[actor] = {
            "blocking_rate": blocking,
            "mean_jct": float(np.mean(jcts)) if jcts else None,
            "episodes": len(done_stats),
            "arrivals_per_episode": float(np.mean(
                [s["num_jobs_arrived"] for s in done_stats])),
            "env_steps_per_sec": steps * args.envs / dt,
        }
        print(f"[1024] {actor}: {json.dumps(results[actor])}", flush=True)

    # pure engine-step throughput at 1024 workers (random valid actions)
    scheds = [drain_episode_schedule(gen, spec, seed=900 + b)
              for b in range(args.envs)]
    eng = GpuEngine(spec, B=args.envs, device=dev, n_jobs_cap=jobs_cap)
    for b in range(args.envs):
        eng.reset_env(b, scheds[b])
    rng = np.random.RandomState(1)
    for _ in range(10):
        eng.step(heuristic_actions("random", eng, scheds, rng))
    torch.cuda.synchronize()
    t2 = time.perf_counter()
    n = 0
    for _ in range(args.bench_steps):
        eng.step(heuristic_actions("random", eng, scheds, rng))
        n += 1
        if bool((eng.T["done"] != 0).any()):
            for b in torch.nonzero(eng.T["done"]).flatten().tolist():
                k = 900 + b + 7777 * n
                scheds[b] = drain_episode_schedule(gen, spec, seed=k)
                eng.reset_env(int(b), scheds[b])
    torch.cuda.synchronize()
    dt = time.perf_counter() - t2
    results["bench"] = {"workers": args.workers, "envs": args.envs,
                        "env_steps_per_sec": n * args.envs / dt,
                        "ms_per_batched_step": dt / n * 1e3}
    print(f"[1024] bench: {json.dumps(results['bench'])}")
    out = os.environ.get("OUT", "gpurun_out/eval_1024.json")
    with open(out, "w") as f:
        json.dump(results, f, indent=2)


if __name__ == "__main__":
    main()
