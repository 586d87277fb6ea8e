"""SRPT op/dep schedulers.

Reference: ``agents/schedulers/srpt_op_scheduler.py:14``,
``srpt_dep_scheduler.py:12``.  Priority convention: ops/deps are ranked by
remaining run time in DESCENDING order and priority = rank index, so the
LARGEST priority value (assigned to the shortest op/dep) is preferred by the
tick loop's argmax.
"""
from __future__ import annotations

from collections import defaultdict
from typing import Dict

import numpy as np

from ..cluster.actions import (DepPlacement, DepSchedule, OpPartition,
                               OpPlacement, OpSchedule)


class SRPTOpScheduler:
    def get(self, op_partition: OpPartition, op_placement: OpPlacement,
            cluster) -> OpSchedule:
        new_placements = op_placement.action
        worker_to_job_to_op_to_priority: Dict = defaultdict(lambda: defaultdict(dict))
        if len(new_placements) == 0:
            return OpSchedule(worker_to_job_to_op_to_priority)

        job_id_to_job = {job_id: job
                         for job_id, job in op_partition.partitioned_jobs.items()
                         if job_id in new_placements}

        # initialise remaining run times from the placement's device types
        worker_to_type = cluster.topology.worker_to_type
        for job_id, job in job_id_to_job.items():
            g = job.graph
            if np.isnan(job.op_remaining[0]):
                for i in range(g.n):
                    wid = new_placements[job_id][g.names[i]]
                    job.reset_op_remaining_run_time(i, device_type=worker_to_type[wid])

        for worker_id, ops in op_placement.worker_to_ops.items():
            job_op_to_cost = {}
            for op in ops:
                job = job_id_to_job[op["job_id"]]
                op_idx = job.graph.name_to_idx[op["op_id"]]
                job_op_to_cost[(op["job_id"], op["op_id"])] = float(
                    job.op_remaining[op_idx])
            # descending cost; priority = rank (longest -> lowest priority 0)
            ranked = sorted(job_op_to_cost, key=job_op_to_cost.get, reverse=True)
            for priority, (job_id, op_name) in enumerate(ranked):
                worker_to_job_to_op_to_priority[worker_id][job_id][op_name] = priority

        return OpSchedule(worker_to_job_to_op_to_priority)


class SRPTDepScheduler:
    def get(self, op_partition: OpPartition, dep_placement: DepPlacement,
            cluster) -> DepSchedule:
        new_placements = dep_placement.action
        channel_to_job_to_dep_to_priority: Dict = defaultdict(lambda: defaultdict(dict))
        if len(new_placements) == 0:
            return DepSchedule(channel_to_job_to_dep_to_priority)

        job_id_to_job = {job_id: job
                         for job_id, job in op_partition.partitioned_jobs.items()
                         if job_id in new_placements}

        for job in job_id_to_job.values():
            if job.graph.m > 0 and np.isnan(job.dep_remaining[0]):
                job.dep_remaining[:] = job.dep_init_run_time

        # cost per placed (job, dep); global descending rank
        jobdep_to_cost = {}
        for job_id, deps in new_placements.items():
            job = job_id_to_job[job_id]
            for dep_idx in deps:
                jobdep_to_cost[(job_id, dep_idx)] = float(job.dep_remaining[dep_idx])
        ranked = sorted(jobdep_to_cost, key=jobdep_to_cost.get, reverse=True)
        for priority, (job_id, dep_idx) in enumerate(ranked):
            for channel_id in dep_placement.job_to_dep_to_channels[job_id][dep_idx]:
                channel_to_job_to_dep_to_priority[channel_id][job_id][dep_idx] = priority

        return DepSchedule(channel_to_job_to_dep_to_priority)
