"""Job-scheduling environment placeholder.

Reference: ``ddls/environments/job_scheduling/job_scheduling_environment.py:6``
is an empty stub (never implemented upstream); kept here so the component
inventory is 1:1 and so a future scheduling-action env has a home.
"""
from __future__ import annotations

class JobSchedulingEnvironment:
    """Not implemented (matches the reference stub)."""

    def __init__(self, *args, **kwargs):
        raise NotImplementedError(
            "JobSchedulingEnvironment is a placeholder (reference "
            "job_scheduling_environment.py is an empty stub); use "
            "RampJobPartitioningEnvironment or JobPlacingAllNodesEnvironment.")
