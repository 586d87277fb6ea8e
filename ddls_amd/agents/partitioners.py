"""Op partitioners (reference ``agents/partitioners/``)."""
from __future__ import annotations

import math
from collections import defaultdict

import numpy as np

from ..cluster.actions import OpPartition
from ..graphs import FWD


def sip_ml_num_partitions(compute_cost: float, min_op_run_time_quantum: float,
                          max_partitions_per_op: int) -> int:
    """SiP-ML rule (reference ``sip_ml_op_partitioner.py:45`` and
    ``ramp_job_partitioning_environment.py:336``):
    clamp_even(ceil(ceil(cost/quantum)/2)*2, max)."""
    return int(max(1, min(
        math.ceil(math.ceil(compute_cost / min_op_run_time_quantum) / 2) * 2,
        max_partitions_per_op)))


class SipMlOpPartitioner:
    """Reference ``sip_ml_op_partitioner.py:8``."""

    def __init__(self, min_op_run_time_quantum: float = 10e-6, **kwargs):
        self.min_op_run_time_quantum = min_op_run_time_quantum

    def get(self, cluster, max_partitions_per_op: int = 2) -> OpPartition:
        if max_partitions_per_op < 1:
            raise ValueError("max_partitions_per_op must be >= 1")
        if max_partitions_per_op > 1 and max_partitions_per_op % 2 != 0:
            raise ValueError("max_partitions_per_op must be even")
        device_type = cluster.device_type
        action = defaultdict(dict)
        for job in cluster.job_queue.jobs.values():
            g = job.graph
            cc = g.compute_cost[device_type]
            for i in range(g.n):
                if g.pass_type[i] != FWD:
                    continue
                num = sip_ml_num_partitions(float(cc[i]),
                                            self.min_op_run_time_quantum,
                                            max_partitions_per_op)
                action[job.job_id][g.names[i]] = num
                action[job.job_id][g.names[int(g.counterpart[i])]] = num
        return OpPartition(dict(action), cluster=cluster)


class RandomOpPartitioner:
    """Random even partition degree per job (reference
    ``random_op_partitioner.py:9``)."""

    def __init__(self, **kwargs):
        pass

    def get(self, cluster, max_partitions_per_op: int = 2) -> OpPartition:
        action = defaultdict(dict)
        for job in cluster.job_queue.jobs.values():
            g = job.graph
            choices = [1] + [d for d in range(2, max_partitions_per_op + 1, 2)]
            num = int(np.random.choice(choices))
            for i in range(g.n):
                if g.pass_type[i] != FWD:
                    continue
                action[job.job_id][g.names[i]] = num
                action[job.job_id][g.names[int(g.counterpart[i])]] = num
        return OpPartition(dict(action), cluster=cluster)
