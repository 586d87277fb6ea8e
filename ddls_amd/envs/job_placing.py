"""Legacy job-placing RL environment: choose HOW MANY workers a job uses.

Reference: ``ddls/environments/job_placing/job_placing_all_nodes_environment.py:19``
(uses the legacy generic ClusterEnvironment; ops are spread sequentially or
randomly over a randomly selected worker set; reward = worker compute
utilisation or (mean/total) JCT; continuous action mode optionally maps a
fraction to a worker count).  ``job_scheduling_environment.py:6`` in the
reference is an empty stub and is intentionally not reproduced.
"""
from __future__ import annotations

import random
from typing import Optional, Union

import numpy as np

from ..cluster.legacy_environment import ClusterEnvironment
from . import spaces


class JobPlacingAllNodesEnvironment:
    def __init__(self,
                 topology_config: dict,
                 node_config: dict,
                 jobs_config: dict,
                 continuous_action_mode: bool = False,
                 worker_selection: str = "random",
                 op_allocation: str = "sequential",
                 reward_function: str = "mean_job_completion_time",
                 max_cluster_simulation_run_time: Union[int, float] = float("inf"),
                 job_queue_capacity: int = 10,
                 name: str = "job_placing_all_nodes"):
        self.jobs_config = jobs_config
        self.continuous_action_mode = continuous_action_mode
        self.worker_selection = worker_selection
        self.op_allocation = op_allocation
        self.reward_function = reward_function
        self.max_cluster_simulation_run_time = max_cluster_simulation_run_time
        self.job_queue_capacity = job_queue_capacity
        self.name = name
        self.cluster = ClusterEnvironment(topology_config=topology_config,
                                          node_config=node_config)
        n_workers = self.cluster.topology.num_workers
        if continuous_action_mode:
            self.action_space = spaces.Box(low=0, high=1, shape=(), dtype=np.float32)
        else:
            self.action_space = spaces.Discrete(n_workers)

    def reset(self, seed: Optional[int] = None):
        self.cluster.reset(jobs_config=self.jobs_config,
                           max_simulation_run_time=self.max_cluster_simulation_run_time,
                           job_queue_capacity=self.job_queue_capacity,
                           seed=seed)
        self._prev_completed_jct_sum = 0.0
        return self._obs()

    def _obs(self):
        """Minimal job encoding: [num ops, num deps, total compute, total mem,
        queue length, num running] (the reference's 358-line observation also
        encodes the padded graph; the PAC-ML successor obs lives in
        envs/observation.py)."""
        if len(self.cluster.job_queue) == 0:
            return np.zeros(6, dtype=np.float32)
        job = next(iter(self.cluster.job_queue.jobs.values()))
        g = job.graph
        dt = self.cluster.device_type
        return np.array([g.n, g.m, float(g.compute_cost[dt].sum()),
                         float(g.memory_cost.sum()),
                         len(self.cluster.job_queue),
                         len(self.cluster.jobs_running)], dtype=np.float32)

    def step(self, action, verbose: bool = False):
        cluster = self.cluster
        n_workers = cluster.topology.num_workers
        if self.continuous_action_mode:
            num = max(1, int(round(float(action) * n_workers)))
        else:
            num = max(1, int(action))
        num = min(num, n_workers)

        placement = {}
        if len(cluster.job_queue) > 0:
            job = next(iter(cluster.job_queue.jobs.values()))
            worker_ids = [w.processor_id for w in cluster.workers]
            if self.worker_selection == "random":
                selected = random.sample(worker_ids, num)
            else:
                selected = worker_ids[:num]
            g = job.graph
            job_placement = {}
            if self.op_allocation == "sequential":
                for i in range(g.n):
                    job_placement[g.names[i]] = selected[i % num]
            else:
                for i in range(g.n):
                    job_placement[g.names[i]] = random.choice(selected)
            placement[job.job_id] = job_placement

        # SRPT job schedule over the placement
        from ..agents.job_managers import SRPTJobScheduler
        schedule = SRPTJobScheduler().get(placement, cluster)
        cluster.step({"job_placement": placement, "job_schedule": schedule})

        reward = self._reward()
        done = cluster.is_done()
        # fast-forward until a job queued or done
        while len(cluster.job_queue) == 0 and not cluster.is_done():
            cluster.step({"job_placement": {}, "job_schedule": {}})
            done = cluster.is_done()
        return self._obs(), reward, done, {}

    def _reward(self):
        stats = self.cluster.episode_stats
        if self.reward_function == "mean_job_completion_time":
            jcts = stats.get("job_completion_time", [])
            return -float(np.mean(jcts)) if jcts else 0.0
        if self.reward_function == "total_job_completion_time":
            jcts = stats.get("job_completion_time", [])
            total = float(np.sum(jcts))
            delta = total - self._prev_completed_jct_sum
            self._prev_completed_jct_sum = total
            return -delta
        if self.reward_function == "worker_compute_utilisation":
            busy = sum(1 for w in self.cluster.workers
                       if len(w.mounted_job_idx_to_ops) > 0)
            return busy / self.cluster.topology.num_workers
        raise ValueError(f"Unrecognised reward_function {self.reward_function}")
