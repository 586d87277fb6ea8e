"""Heuristic baseline actors (the PAC-ML paper baselines).

Reference: ``ddls/environments/ramp_job_partitioning/agents/`` — ``random.py:3``,
``no_parallelism.py:1``, ``min_parallelism.py:1``, ``max_parallelism.py:1``,
``sip_ml.py:4``, ``acceptable_jct.py:3``.
"""
from __future__ import annotations

import math
from typing import Optional

import numpy as np


class Random:
    def __init__(self, name: str = "random", **kwargs):
        self.name = name

    def compute_action(self, obs, *args, **kwargs):
        valid = obs["action_set"][obs["action_mask"].astype(bool)]
        if len(valid) > 1:
            return int(np.random.choice(valid[1:]))
        return int(valid[0])


class NoParallelism:
    def __init__(self, name: str = "no_parallelism", **kwargs):
        self.name = name

    def compute_action(self, obs, *args, **kwargs):
        valid = obs["action_set"][obs["action_mask"].astype(bool)]
        return 1 if len(valid) > 1 else 0


class MinParallelism:
    def __init__(self, name: str = "min_parallelism", **kwargs):
        self.name = name

    def compute_action(self, obs, *args, **kwargs):
        valid = obs["action_set"][obs["action_mask"].astype(bool)]
        if len(valid) > 2:
            return 2
        if len(valid) == 2:
            return 1
        return 0


class MaxParallelism:
    def __init__(self, name: str = "max_parallelism", **kwargs):
        self.name = name

    def compute_action(self, obs, *args, **kwargs):
        valid = obs["action_set"][obs["action_mask"].astype(bool)]
        if len(valid) > 1:
            return int(valid[1:][-1])
        return int(valid[0])


class SiPML:
    """Always partition up to a statically fixed max (reference ``sip_ml.py:4``)."""

    def __init__(self, max_partitions_per_op: Optional[int] = None,
                 name: str = "sip_ml", **kwargs):
        self.max_partitions_per_op = max_partitions_per_op
        self.name = name

    def compute_action(self, obs, *args, **kwargs):
        valid = obs["action_set"][obs["action_mask"].astype(bool)]
        if len(valid) > 1:
            max_allowed = int(valid[-1])
            if self.max_partitions_per_op is not None:
                return min(self.max_partitions_per_op, max_allowed)
            return max_allowed
        return int(valid[0])


class AcceptableJCT:
    """Smallest valid partition degree whose parallelism approximately meets
    the job's max-acceptable-JCT contract (reference ``acceptable_jct.py:26-40``)."""

    def __init__(self, name: str = "acceptable_jct", **kwargs):
        self.name = name

    def compute_action(self, obs, job_to_place=None, *args, **kwargs):
        valid = obs["action_set"][obs["action_mask"].astype(bool)]
        if len(valid) <= 1:
            return int(valid[0])
        dt = next(iter(job_to_place.details["job_sequential_completion_time"].keys()))
        acceptable = int(math.ceil(
            job_to_place.details["job_sequential_completion_time"][dt]
            / job_to_place.details["max_acceptable_job_completion_time"][dt]))
        action = int(valid[-1])
        for a in valid:
            if a == acceptable or a > acceptable:
                action = int(a)
                break
        return action


ACTORS = {
    "random": Random,
    "no_parallelism": NoParallelism,
    "min_parallelism": MinParallelism,
    "max_parallelism": MaxParallelism,
    "sip_ml": SiPML,
    "acceptable_jct": AcceptableJCT,
}
