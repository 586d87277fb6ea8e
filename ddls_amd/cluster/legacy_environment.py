"""Legacy generic cluster environment: true per-tick dynamic simulation.

Reference: ``ddls/environments/cluster/cluster_environment.py:28`` (the
pre-RAMP simulator kept alongside the RAMP env).  Jobs are placed at the
job level (one placement dict per job), multiple jobs may share workers, and
mounted ops execute tick by tick ("TEMPORARY: Assume no network communication
overhead" — child dependencies are satisfied the instant their parent op
completes, ``:286``).
"""
from __future__ import annotations

from collections import defaultdict
from typing import Dict, Optional, Union

import numpy as np

from ..jobs import Job, JobQueue, JobsGenerator
from ..topology import build_topology
from ..utils import Stopwatch, seed_everything
from .environment import RampClusterEnvironment


class ClusterEnvironment:
    def __init__(self, topology_config: dict, node_config: dict,
                 name: str = "cluster", suppress_warnings: bool = True):
        self.topology_config = topology_config
        self.node_config = node_config
        self.name = name
        self.topology = build_topology(topology_config)
        # reuse the RAMP populate logic for workers
        RampClusterEnvironment._populate_topology(self, self.topology, node_config)
        self.stopwatch = Stopwatch()
        self.reset_counter = 0

    @property
    def device_type(self):
        return next(iter(self.topology.worker_types))

    def reset(self, jobs_config: dict,
              max_simulation_run_time: Union[int, float] = float("inf"),
              job_queue_capacity: int = 10,
              seed: Optional[int] = None):
        self.reset_counter += 1
        if seed is not None:
            seed_everything(seed)
        self.stopwatch.reset()
        self.jobs_generator = JobsGenerator(**jobs_config)
        self.max_simulation_run_time = max_simulation_run_time
        self.job_queue = JobQueue(queue_capacity=job_queue_capacity)
        self.num_jobs_arrived = 0
        self.jobs_running: Dict[int, Job] = {}
        self.jobs_completed: Dict[int, Job] = {}
        self.jobs_blocked: Dict[int, Job] = {}
        self.job_op_to_worker = {}
        self.job_op_priority = {}  # (job_idx, op_idx) -> priority
        self.job_idx_to_job_id = {}
        self.job_id_to_job_idx = {}
        self.step_counter = 0
        self.steps_log = defaultdict(list)
        self.episode_stats = defaultdict(list)
        self.episode_stats["num_jobs_arrived"] = 0
        self.episode_stats["num_jobs_completed"] = 0
        self.episode_stats["num_jobs_blocked"] = 0
        for workers in self.topology.node_workers:
            for w in workers.values():
                w.reset()
        self.time_next_job_to_arrive = 0.0
        self.job_queue.add(self._get_next_job())
        return None

    def _get_next_job(self) -> Job:
        job = self.jobs_generator.sample_job()
        job_idx = self.num_jobs_arrived
        job.original_job.job_id = job.job_id
        job.original_job.details["job_idx"] = job_idx
        job.register_job_arrived(time_arrived=self.stopwatch.time(),
                                 job_idx=job_idx)
        self.time_next_job_to_arrive += self.jobs_generator.sample_interarrival_time()
        self.job_idx_to_job_id[job_idx] = job.job_id
        self.job_id_to_job_idx[job.job_id] = job_idx
        self.num_jobs_arrived += 1
        self.episode_stats["num_jobs_arrived"] += 1
        return job

    # ------------------------------------------------------------------
    def _place_jobs(self, job_placement: Dict):
        """job_placement: job_id -> op_name -> worker_id."""
        for job_id, placement in job_placement.items():
            job = self.job_queue.jobs[job_id]
            g = job.graph
            job_idx = job.details["job_idx"]
            job.op_worker = np.full(g.n, -1, dtype=np.int64)
            for op_name, worker_id in placement.items():
                op_idx = g.name_to_idx[str(op_name)]
                node = self.topology.worker_to_node[worker_id]
                worker = self.topology.node_workers[node][worker_id]
                worker.mount(job=job, op_idx=op_idx)
                job.details["mounted_workers"].add(worker_id)
                job.reset_op_remaining_run_time(op_idx, worker.device_type)
                self.job_op_to_worker[(job_idx, op_idx)] = worker_id
                job.op_worker[op_idx] = self.worker_id_to_index[worker_id]
            job.register_job_running(time_started=self.stopwatch.time())
            self.jobs_running[job_idx] = job
            self.job_queue.remove(job)

    def _schedule_jobs(self, job_schedule: Dict):
        """job_schedule: worker_id -> job_id -> op_name -> priority."""
        for worker_id, job_to_ops in job_schedule.items():
            for job_id, op_to_priority in job_to_ops.items():
                job_idx = self.job_id_to_job_idx[job_id]
                job = self.jobs_running.get(job_idx)
                if job is None:
                    continue
                for op_name, priority in op_to_priority.items():
                    op_idx = job.graph.name_to_idx[str(op_name)]
                    self.job_op_priority[(job_idx, op_idx)] = priority

    def _priority_op_per_worker(self):
        """Highest-priority ready op per worker over ALL running jobs."""
        out = {}
        for w_idx, worker in enumerate(self.workers):
            best, best_p = None, None
            for job_idx in sorted(worker.mounted_job_idx_to_ops.keys()):
                job = self.jobs_running.get(job_idx)
                if job is None:
                    continue
                for op_idx in worker.mounted_job_idx_to_ops[job_idx]:
                    if job.ops_ready[op_idx]:
                        p = self.job_op_priority.get((job_idx, op_idx), 0)
                        if best is None or p > best_p:
                            best, best_p = (job_idx, op_idx), p
            if best is not None:
                out[w_idx] = best
        return out

    def _tick_workers(self, max_tick=None):
        priority_ops = self._priority_op_per_worker()
        shortest = float("inf")
        for job_idx, op_idx in priority_ops.values():
            rem = self.jobs_running[job_idx].op_remaining[op_idx]
            shortest = min(shortest, rem)
        tick = min(shortest, max_tick) if max_tick is not None else shortest
        if np.isinf(tick):
            tick = max_tick if max_tick is not None else 0.0
        completed = defaultdict(list)
        for job_idx, op_idx in priority_ops.values():
            job = self.jobs_running[job_idx]
            job.tick_op(op_idx, tick=tick)
            if job.ops_completed[op_idx]:
                completed[job_idx].append(op_idx)
        self.stopwatch.tick(tick)
        return completed

    # ------------------------------------------------------------------
    def step(self, actions: Dict, verbose: bool = False):
        self.step_stats = {"step_start_time": self.stopwatch.time(),
                           "num_jobs_arrived": 0, "num_jobs_completed": 0,
                           "num_jobs_blocked": 0}
        self._place_jobs(actions.get("job_placement", {}))
        self._schedule_jobs(actions.get("job_schedule", {}))

        step_done = False
        while not step_done:
            max_tick = min(self.time_next_job_to_arrive - self.stopwatch.time(),
                           self.max_simulation_run_time - self.stopwatch.time())
            completed = self._tick_workers(max_tick=max_tick)

            # no network model: deps complete the moment the parent op does
            for job_idx, op_idxs in completed.items():
                job = self.jobs_running[job_idx]
                for op_idx in op_idxs:
                    for e in job.graph.out_edges_of(op_idx):
                        if job.deps_ready[e]:
                            job.dep_remaining[e] = 0.0
                            job._register_completed_dep(int(e))

            for job_idx in list(completed.keys()):
                job = self.jobs_running[job_idx]
                if job.is_training_step_complete():
                    job.reset_job_training_step(preserve_mounts=True)
                if job.is_job_complete():
                    self._register_completed_job(job)
                    step_done = True

            if len(self.jobs_generator) > 0:
                if self.stopwatch.time() >= self.time_next_job_to_arrive:
                    next_job = self._get_next_job()
                    self.step_stats["num_jobs_arrived"] += 1
                    if self.job_queue.can_fit(next_job):
                        self.job_queue.add(next_job)
                    else:
                        self._register_blocked_job(next_job)
                    step_done = True
            else:
                self.time_next_job_to_arrive = float("inf")

            if self.is_done():
                step_done = True

        self.step_stats["step_end_time"] = self.stopwatch.time()
        self.step_stats["job_queue_length"] = len(self.job_queue)
        for k, v in self.step_stats.items():
            self.steps_log[k].append(v)
        self.step_counter += 1
        return None, None, None, self.is_done(), None

    def _register_completed_job(self, job: Job):
        job.register_job_completed(time_completed=self.stopwatch.time())
        job_idx = job.details["job_idx"]
        self.jobs_completed[job_idx] = job
        self.step_stats["num_jobs_completed"] += 1
        self.episode_stats["num_jobs_completed"] += 1
        self.episode_stats["job_completion_time"].append(
            job.details["time_completed"] - job.details["time_arrived"])
        # unmount
        for op_idx in np.flatnonzero(job.op_worker >= 0):
            key = (job_idx, int(op_idx))
            worker_id = self.job_op_to_worker.pop(key, None)
            if worker_id is not None:
                node = self.topology.worker_to_node[worker_id]
                self.topology.node_workers[node][worker_id].unmount(job, int(op_idx))
        del self.jobs_running[job_idx]

    def _register_blocked_job(self, job: Job):
        self.jobs_blocked[job.details["job_idx"]] = job
        self.step_stats["num_jobs_blocked"] += 1
        self.episode_stats["num_jobs_blocked"] += 1
        self.job_queue.remove(job)

    def is_done(self, verbose: bool = False) -> bool:
        if self.max_simulation_run_time is not None:
            if self.stopwatch.time() >= self.max_simulation_run_time:
                return True
        return (len(self.jobs_generator) == 0 and len(self.jobs_running) == 0
                and len(self.job_queue) == 0)
