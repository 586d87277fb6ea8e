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


class SRPTJobPrioritiser:
    """Order jobs by total size ascending (reference
    ``managers/prioritisers/srpt_job_prioritiser.py``)."""

    def prioritise(self, jobs):
        sizes = np.array([j.job_total_operation_memory_cost
                          + j.job_total_dependency_size for j in jobs])
        return [jobs[i] for i in np.argsort(sizes)]


class RandomJobPartitioner:
    """Job-level random partitioner for the legacy path (reference
    ``managers/partitioners/random_job_partitioner.py``): picks an even
    partition degree per job and returns the degree map (the RAMP-side
    graph transform lives in cluster/partition.py)."""

    def __init__(self, max_partitions_per_op: int = 2):
        self.max_partitions_per_op = max_partitions_per_op

    def get(self, cluster):
        choices = [1] + list(range(2, self.max_partitions_per_op + 1, 2))
        return {job_id: int(np.random.choice(choices))
                for job_id in cluster.job_queue.jobs}


class AllReduceJobCommunicator:
    """Gradient-sync communicator for the legacy cluster.

    The reference's ``managers/communicators/all_reduce_job_communicator.py``
    raises NotImplemented; here it is functional: prices a per-training-step
    ring all-reduce of the job's parameter bytes over the workers the job is
    mounted on, using the analytic collective model (xGMI-profiled by
    default: 7 p2p links, per-link bound ring -> 2(n-1)/n * S over one link).
    """

    def __init__(self, link_bandwidth: float = 153e9, num_links: int = 7,
                 latency: float = 1e-6):
        self.link_bandwidth = link_bandwidth
        self.num_links = num_links
        self.latency = latency

    def all_reduce_time(self, message_size: float, num_workers: int) -> float:
        if num_workers <= 1:
            return 0.0
        # ring all-reduce on point-to-point links is per-link bound
        per_link_bytes = 2.0 * (num_workers - 1) / num_workers * message_size
        steps = 2 * (num_workers - 1)
        return steps * self.latency + per_link_bytes / self.link_bandwidth

    def communicate(self, job, cluster) -> float:
        _jmap = cluster.job_op_to_worker.get(job.details["job_idx"], {})
        workers = {_jmap.get(i)
                   for i in range(job.graph.n)}
        workers.discard(None)
        message = float(job.graph.memory_cost.sum())
        return self.all_reduce_time(message, len(workers))
