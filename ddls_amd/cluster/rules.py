"""RAMP validity rules (reference ``ramp_rules.py:2-40``)."""
from __future__ import annotations


def check_if_ramp_op_placement_rules_broken(worker, job) -> list:
    """Rule: no worker can have ops from more than one job."""
    rules_broken = []
    job_idx = job.details["job_idx"]
    if job_idx not in worker.mounted_job_idx_to_ops:
        if len(worker.mounted_job_idx_to_ops) > 0:
            rules_broken.append("one_job_per_worker")
    return rules_broken


def check_if_ramp_dep_placement_rules_broken(channel, job) -> list:
    """Rule: no channel can have flows from more than one job."""
    rules_broken = []
    job_idx = job.details["job_idx"]
    if job_idx not in channel.mounted_job_idx_to_deps:
        if len(channel.mounted_job_idx_to_deps) > 0:
            rules_broken.append("one_job_per_channel")
    return rules_broken
