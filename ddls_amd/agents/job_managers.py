"""Job-level placers/schedulers for the legacy generic cluster.

Reference: ``ddls/managers/`` — ``placers/random_job_placer.py:21``,
``schedulers/{srpt,fifo,random}_job_scheduler.py``,
``prioritisers/srpt_job_prioritiser.py`` (used by ``scripts/run_sim.py:53-58``).
"""
from __future__ import annotations

import random
from collections import defaultdict
from typing import Dict

import numpy as np


class RandomJobPlacer:
    """Place every op of each queued job on a random memory-feasible worker."""

    def get(self, cluster) -> Dict:
        placement = {}
        free_mem = {w.processor_id: w.memory_capacity - w.memory_occupied
                    for w in cluster.workers}
        for job_id, job in cluster.job_queue.jobs.items():
            g = job.graph
            job_placement = {}
            ok = True
            for i in range(g.n):
                candidates = [wid for wid, mem in free_mem.items()
                              if mem >= g.memory_cost[i]]
                if not candidates:
                    ok = False
                    break
                wid = random.choice(candidates)
                job_placement[g.names[i]] = wid
                free_mem[wid] -= g.memory_cost[i]
            if ok:
                placement[job_id] = job_placement
        return placement


class _BaseJobScheduler:
    def _rank(self, costs: Dict) -> Dict:
        """costs: (job_id, op_name) -> cost; returns priorities (higher value
        = preferred), longest-cost op lowest priority (SRPT convention)."""
        ranked = sorted(costs, key=costs.get, reverse=True)
        return {key: p for p, key in enumerate(ranked)}


class SRPTJobScheduler(_BaseJobScheduler):
    """Shortest-remaining-processing-time over jobs: ops of the job with the
    least remaining work get the highest priorities."""

    def get(self, placement: Dict, cluster) -> Dict:
        schedule = defaultdict(lambda: defaultdict(dict))
        dt = cluster.device_type
        job_costs = {}
        for job_id in placement:
            job = cluster.job_queue.jobs.get(job_id)
            if job is None:
                continue
            job_costs[job_id] = (job.details["job_sequential_completion_time"][dt]
                                 * 1.0)
        ranked_jobs = sorted(job_costs, key=job_costs.get, reverse=True)
        job_priority = {jid: p for p, jid in enumerate(ranked_jobs)}
        for job_id, job_placement in placement.items():
            job = cluster.job_queue.jobs.get(job_id)
            if job is None:
                continue
            g = job.graph
            for op_name, worker_id in job_placement.items():
                op_idx = g.name_to_idx[str(op_name)]
                cost = float(g.compute_cost[dt][op_idx])
                # higher job priority dominates; within a job, shorter op first
                schedule[worker_id][job_id][op_name] = (
                    job_priority[job_id] * 1000000 - cost)
        return {w: {j: dict(ops) for j, ops in jobs.items()}
                for w, jobs in schedule.items()}


class FIFOJobScheduler(_BaseJobScheduler):
    """First-in-first-out: earlier-arrived jobs get higher priority."""

    def get(self, placement: Dict, cluster) -> Dict:
        schedule = defaultdict(lambda: defaultdict(dict))
        for job_id, job_placement in placement.items():
            job = cluster.job_queue.jobs.get(job_id)
            if job is None:
                continue
            prio = -job.details["job_idx"]  # earlier arrival -> higher
            for op_name, worker_id in job_placement.items():
                schedule[worker_id][job_id][op_name] = prio
        return {w: {j: dict(ops) for j, ops in jobs.items()}
                for w, jobs in schedule.items()}


class RandomJobScheduler(_BaseJobScheduler):
    def get(self, placement: Dict, cluster) -> Dict:
        schedule = defaultdict(lambda: defaultdict(dict))
        for job_id, job_placement in placement.items():
            for op_name, worker_id in job_placement.items():
                schedule[worker_id][job_id][op_name] = float(np.random.rand())
        return {w: {j: dict(ops) for j, ops in jobs.items()}
                for w, jobs in schedule.items()}


JOB_PLACERS = {"random": RandomJobPlacer}
JOB_SCHEDULERS = {"srpt": SRPTJobScheduler, "fifo": FIFOJobScheduler,
                  "random": RandomJobScheduler}
