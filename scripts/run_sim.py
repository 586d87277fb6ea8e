#!/usr/bin/env python3
"""Run the legacy generic cluster simulation with job-level heuristics
(reference ``scripts/run_sim.py:71``)."""
import argparse
import json
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np

from ddls_amd.agents.job_managers import JOB_PLACERS, JOB_SCHEDULERS
from ddls_amd.cluster.legacy_environment import ClusterEnvironment
from ddls_amd.utils import seed_everything
from ddls_amd.workloads import ensure_default_set


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--placer", default="random", choices=list(JOB_PLACERS))
    ap.add_argument("--scheduler", default="srpt", choices=list(JOB_SCHEDULERS))
    ap.add_argument("--num-jobs", type=int, default=20)
    ap.add_argument("--workers", type=int, default=16)
    ap.add_argument("--seed", type=int, default=0)
    args = ap.parse_args()

    seed_everything(args.seed)
    env = ClusterEnvironment(
        topology_config={"type": "torus", "kwargs": {
            "x_dims": args.workers, "y_dims": 1, "z_dims": 1,
            "num_channels": 1}},
        node_config={"type_1": {"num_nodes": args.workers, "workers_config": [
            {"num_workers": 1, "worker": "ddls_amd.devices.MI355X"}]}})
    env.reset(jobs_config={
        "path_to_files": ensure_default_set(),
        "replication_factor": max(1, args.num_jobs // 5),
        "job_sampling_mode": "remove",
        "job_interarrival_time_dist": {
            "_target_": "ddls_amd.distributions.Fixed", "val": 100},
        "num_training_steps": 2,
        "processor_type_profiled": "MI355X",
    }, max_simulation_run_time=1e6, seed=args.seed)

    placer = JOB_PLACERS[args.placer]()
    scheduler = JOB_SCHEDULERS[args.scheduler]()
    done = False
    while not done:
        placement = placer.get(env)
        schedule = scheduler.get(placement, env)
        _, _, _, done, _ = env.step({"job_placement": placement,
                                     "job_schedule": schedule})
    stats = env.episode_stats
    print(json.dumps({
        "num_jobs_arrived": stats["num_jobs_arrived"],
        "num_jobs_completed": stats["num_jobs_completed"],
        "num_jobs_blocked": stats["num_jobs_blocked"],
        "mean_job_completion_time": float(np.mean(stats["job_completion_time"]))
        if stats["job_completion_time"] else None,
        "sim_time": env.stopwatch.time(),
    }))


if __name__ == "__main__":
    main()
