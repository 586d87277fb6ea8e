"""Reward functions for the partitioning env.

Reference: ``ddls/environments/ramp_job_partitioning/rewards/`` —
``lookahead_job_completion_time.py:9``, ``job_acceptance.py:9``,
``mean_compute_throughput.py:9``, ``mean_cluster_throughput.py``,
``mean_demand_total_throughput.py``, ``multi_objective_jct_blocking.py:9``.
"""
from __future__ import annotations

import math
from typing import Union


from .base import DDLSRewardFunction


class RewardFunction(DDLSRewardFunction):
    def reset(self, env=None, **kwargs):
        pass

    def extract(self, env, done: bool):
        raise NotImplementedError


class LookaheadJobCompletionTime(RewardFunction):
    """-(lookahead JCT); blocked jobs get fail_reward (default: the job's
    sequential JCT) x fail_reward_factor; optional 1/x, log10, normalisation."""

    def __init__(self,
                 fail_reward: Union[int, float, str] = "job_sequential_completion_time",
                 fail_reward_factor: float = 1,
                 sign: int = -1,
                 inverse: bool = False,
                 transform_with_log: bool = False,
                 normaliser: Union[str, None] = None):
        self.fail_reward = fail_reward
        self.fail_reward_factor = fail_reward_factor
        self.sign = sign
        self.inverse = inverse
        self.transform_with_log = transform_with_log
        self.normaliser = normaliser

    def _normalise(self, reward, job, env):
        dt = env.cluster.device_type
        if self.normaliser == "job_sequential_completion_time":
            return reward / job.details["job_sequential_completion_time"][dt]
        if self.normaliser == "job_sequential_completion_time_times_fail_reward_factor":
            return reward / (job.details["job_sequential_completion_time"][dt]
                             * self.fail_reward_factor)
        raise ValueError(f"Unrecognised normaliser {self.normaliser}")

    def extract(self, env, done: bool):
        job_idx = env.last_job_arrived_job_idx
        if job_idx in env.placed_job_idxs:
            if job_idx in env.cluster.jobs_running:
                job = env.cluster.jobs_running[job_idx]
            elif job_idx in env.cluster.jobs_completed:
                job = env.cluster.jobs_completed[job_idx]
            else:
                raise KeyError(f"job_idx {job_idx} not in running or completed")
            reward = job.details["lookahead_job_completion_time"]
            if self.normaliser is not None and reward != 0:
                reward = self._normalise(reward, job, env)
        else:
            job = env.cluster.jobs_blocked[job_idx]
            if isinstance(self.fail_reward, (int, float)):
                reward = self.fail_reward * self.fail_reward_factor
            elif self.fail_reward == "job_sequential_completion_time":
                dt = env.cluster.device_type
                reward = (job.details["job_sequential_completion_time"][dt]
                          * self.fail_reward_factor)
            else:
                raise ValueError(f"Unrecognised fail_reward {self.fail_reward}")
            if self.normaliser is not None and reward != 0:
                reward = self._normalise(reward, job, env)

        if self.inverse and reward != 0:
            reward = 1 / reward
        reward *= self.sign
        if self.transform_with_log:
            sign = math.copysign(1, reward)
            reward = sign * math.log(1 + abs(reward), 10)
        return reward


class JobAcceptance(RewardFunction):
    """+success_reward if placed else fail_reward (reference
    ``job_acceptance.py:9-33``)."""

    def __init__(self, fail_reward: float = -1, success_reward: float = 1):
        self.fail_reward = fail_reward
        self.success_reward = success_reward

    def extract(self, env, done: bool):
        if env.last_job_arrived_job_idx in env.placed_job_idxs:
            return self.success_reward
        return self.fail_reward


class MeanComputeThroughput(RewardFunction):
    def extract(self, env, done: bool):
        vals = [s.get("mean_compute_throughput", 0)
                for s in env.cluster_step_stats.values()]
        return float(sum(vals) / len(vals)) if vals else 0.0


class MeanClusterThroughput(RewardFunction):
    def extract(self, env, done: bool):
        vals = [s.get("mean_cluster_throughput", 0)
                for s in env.cluster_step_stats.values()]
        return float(sum(vals) / len(vals)) if vals else 0.0


class MeanDemandTotalThroughput(RewardFunction):
    def extract(self, env, done: bool):
        vals = [s.get("mean_demand_total_throughput", 0)
                for s in env.cluster_step_stats.values()]
        return float(sum(vals) / len(vals)) if vals else 0.0


class MultiObjectiveJCTBlocking(RewardFunction):
    """Weighted combination of -JCT and acceptance
    (reference ``multi_objective_jct_blocking.py:9-90``)."""

    def __init__(self, jct_weight: float = 1.0, blocking_weight: float = 1.0,
                 **jct_kwargs):
        self.jct = LookaheadJobCompletionTime(**jct_kwargs)
        self.acceptance = JobAcceptance()
        self.jct_weight = jct_weight
        self.blocking_weight = blocking_weight

    def reset(self, env=None, **kwargs):
        self.jct.reset(env)
        self.acceptance.reset(env)

    def extract(self, env, done: bool):
        return (self.jct_weight * self.jct.extract(env, done)
                + self.blocking_weight * self.acceptance.extract(env, done))


REWARD_FUNCTIONS = {
    "lookahead_job_completion_time": LookaheadJobCompletionTime,
    "job_acceptance": JobAcceptance,
    "mean_compute_throughput": MeanComputeThroughput,
    "mean_cluster_throughput": MeanClusterThroughput,
    "mean_demand_total_throughput": MeanDemandTotalThroughput,
    "multi_objective_jct_blocking": MultiObjectiveJCTBlocking,
}
