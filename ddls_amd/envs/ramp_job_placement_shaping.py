"""Placement-shaping RL environment: the agent picks the (c, r, s) meta-block
shape per job; partitioning is done by an internal heuristic partitioner.

Reference: ``ddls/environments/ramp_job_placement_shaping/
ramp_job_placement_shaping_environment.py:42`` (the pre-PAC-ML env variant)
plus the job placement shapers
(``agents/job_placement_shapers/ramp_{random,first_fit}_...py:10``).

NB the reference's shaping env passes ``job_placement_shape`` into an Action
whose constructor no longer accepts it (``actions/action.py:3``) — a dead
parameter in the reference; here the shape is honoured by the placer and the
Action treats it as a sixth handled-job constraint.
"""
from __future__ import annotations

from typing import Optional, Union

import numpy as np

from ..agents.partitioners import RandomOpPartitioner, SipMlOpPartitioner
from ..agents.placement_utils import check_meta_block_valid, dummy_ramp
from ..agents.placers import FirstFitDepPlacer, RampFirstFitOpPlacer
from ..agents.schedulers import SRPTDepScheduler, SRPTOpScheduler
from ..cluster.actions import Action, JobPlacementShape, OpPartition
from ..cluster.environment import RampClusterEnvironment
from . import spaces
from .observation import RampJobPartitioningObservation
from .rewards import REWARD_FUNCTIONS


class RampJobPlacementShapingObservation(RampJobPartitioningObservation):
    """Same graph/feature encoding as the partitioning observation, but the
    action set/mask enumerate meta-block shapes (c, r, s)
    (reference ``ramp_job_placement_shaping_observation.py``)."""

    def get_action_set_and_action_mask(self, env):
        cluster = env.cluster
        ramp_shape = cluster.topology.shape
        ramp_topology = dummy_ramp(ramp_shape, cluster)
        job = next(iter(cluster.job_queue.jobs.values()))
        degree = env.op_partition.job_id_to_max_partition_degree.get(job.job_id, 1) \
            if env.op_partition is not None else 1
        num_avail = cluster.topology.num_workers - len(cluster.mounted_workers)
        action_set, action_mask = [0], [True]
        action = 1
        for c in range(1, ramp_shape[0] + 1):
            for r in range(1, ramp_shape[1] + 1):
                for s in range(1, ramp_shape[2] + 1):
                    action_set.append(action)
                    action_mask.append(check_meta_block_valid(
                        c, r, s, ramp_topology, ramp_shape, degree, num_avail))
                    action += 1
        return action_set, action_mask


class RampJobPlacementShapingEnvironment:
    def __init__(self,
                 topology_config: dict,
                 node_config: dict,
                 jobs_config: dict,
                 op_partitioner: str = "sip_ml_op_partitioner",
                 op_partitioner_kwargs: Optional[dict] = None,
                 reward_function: str = "lookahead_job_completion_time",
                 reward_function_kwargs: Optional[dict] = None,
                 pad_obs_kwargs: Optional[dict] = None,
                 max_simulation_run_time: Union[int, float, None] = None,
                 job_queue_capacity: int = 10,
                 name: str = "ramp_job_placement_shaping",
                 suppress_warnings: bool = True):
        self.topology_config = topology_config
        self.node_config = node_config
        self.jobs_config = jobs_config
        self.max_simulation_run_time = (float("inf") if max_simulation_run_time
                                        is None else max_simulation_run_time)
        self.job_queue_capacity = job_queue_capacity
        self.name = name

        self.cluster = RampClusterEnvironment(
            topology_config=topology_config, node_config=node_config,
            suppress_warnings=suppress_warnings)

        shape = self.cluster.topology.shape
        self.action_space = spaces.Discrete(shape[0] * shape[1] * shape[2] + 1)
        self.action_to_job_placement_shape = {0: None}
        a = 1
        for c in range(1, shape[0] + 1):
            for r in range(1, shape[1] + 1):
                for s in range(1, shape[2] + 1):
                    self.action_to_job_placement_shape[a] = (c, r, s)
                    a += 1

        if op_partitioner == "sip_ml_op_partitioner":
            self.op_partitioner = SipMlOpPartitioner(**(op_partitioner_kwargs or {}))
        elif op_partitioner == "random_op_partitioner":
            self.op_partitioner = RandomOpPartitioner(**(op_partitioner_kwargs or {}))
        else:
            raise ValueError(f"Unrecognised op_partitioner {op_partitioner}")
        self.op_placer = RampFirstFitOpPlacer()
        self.op_scheduler = SRPTOpScheduler()
        self.dep_placer = FirstFitDepPlacer()
        self.dep_scheduler = SRPTDepScheduler()

        self.observation_function = RampJobPlacementShapingObservation(
            max_partitions_per_op=1, pad_obs_kwargs=pad_obs_kwargs)
        self.reward_function = REWARD_FUNCTIONS[reward_function](
            **(reward_function_kwargs or {}))
        self.op_partition = None

    def reset(self, seed: Optional[int] = None, verbose: bool = False):
        self.step_counter = 0
        self.cluster.reset(jobs_config=self.jobs_config,
                           max_simulation_run_time=self.max_simulation_run_time,
                           job_queue_capacity=self.job_queue_capacity,
                           seed=seed)
        max_partitions = self.cluster.jobs_generator.max_partitions_per_op_in_observation
        self.op_partition = self.op_partitioner.get(
            cluster=self.cluster, max_partitions_per_op=max_partitions)
        self.obs = self.observation_function.reset(self)
        self.observation_space = self.observation_function.observation_space
        self.reward_function.reset(self.cluster)
        return self.obs

    def _is_done(self):
        return self.cluster.is_done()

    def step(self, action: int, verbose: bool = False):
        if not self.obs["action_mask"][action]:
            raise ValueError(f"action {action} invalid given mask "
                             f"{self.obs['action_mask']}")
        shape = self.action_to_job_placement_shape[int(action)]
        if shape is not None:
            job_id = next(iter(self.op_partition.job_ids))
            self.job_placement_shape = JobPlacementShape({job_id: tuple(shape)})
        else:
            self.job_placement_shape = JobPlacementShape({})

        self.op_placement = self.op_placer.get(
            op_partition=self.op_partition, cluster=self.cluster,
            job_placement_shape=self.job_placement_shape)
        self.op_schedule = self.op_scheduler.get(
            op_partition=self.op_partition, op_placement=self.op_placement,
            cluster=self.cluster)
        self.dep_placement = self.dep_placer.get(
            op_partition=self.op_partition, op_placement=self.op_placement,
            cluster=self.cluster)
        self.dep_schedule = self.dep_scheduler.get(
            op_partition=self.op_partition, dep_placement=self.dep_placement,
            cluster=self.cluster)
        self.action = Action(op_partition=self.op_partition,
                             op_placement=self.op_placement,
                             op_schedule=self.op_schedule,
                             dep_placement=self.dep_placement,
                             dep_schedule=self.dep_schedule,
                             job_placement_shape=self.job_placement_shape)
        self.placed_job_idxs = set(self.action.job_idxs)
        self.last_job_arrived_job_idx = self.cluster.last_job_arrived_job_idx

        self.cluster.step(self.action)
        for job_idx in list(self.placed_job_idxs):
            if job_idx in self.cluster.jobs_blocked:
                self.placed_job_idxs.remove(job_idx)
        self.reward = self.reward_function.extract(env=self, done=self._is_done())

        while len(self.cluster.job_queue) == 0 and not self.cluster.is_done():
            self.cluster.step(Action())

        self.done = self._is_done()
        if not self.done:
            max_partitions = (self.cluster.jobs_generator
                              .max_partitions_per_op_in_observation)
            self.op_partition = self.op_partitioner.get(
                cluster=self.cluster, max_partitions_per_op=max_partitions)
            self.obs = self.observation_function.extract(env=self, done=False)
        self.step_counter += 1
        return self.obs, self.reward, self.done, {}


class FirstFitJobPlacementShaper:
    """Pick the FIRST valid meta-block shape (reference
    ``ramp_job_placement_shaping/agents/ramp_first_fit_job_placement_shaper.py:10``)."""

    def compute_action(self, obs, **kwargs):
        mask = np.asarray(obs["action_mask"]).astype(bool)
        action_set = np.asarray(obs["action_set"])
        valid = action_set[mask]
        nonzero = valid[valid != 0]
        return int(nonzero[0]) if len(nonzero) else 0


class LastFitJobPlacementShaper:
    """Pick the LAST valid meta-block shape."""

    def compute_action(self, obs, **kwargs):
        mask = np.asarray(obs["action_mask"]).astype(bool)
        action_set = np.asarray(obs["action_set"])
        valid = action_set[mask]
        nonzero = valid[valid != 0]
        return int(nonzero[-1]) if len(nonzero) else 0


class RandomJobPlacementShaper:
    """Uniform over valid shapes (reference
    ``agents/ramp_random_job_placement_shaper.py:10``)."""

    def compute_action(self, obs, **kwargs):
        mask = np.asarray(obs["action_mask"]).astype(bool)
        action_set = np.asarray(obs["action_set"])
        valid = action_set[mask]
        nonzero = valid[valid != 0]
        if len(nonzero) == 0:
            return 0
        return int(np.random.choice(nonzero))


SHAPING_AGENTS = {
    "first_fit": FirstFitJobPlacementShaper,
    "last_fit": LastFitJobPlacementShaper,
    "random": RandomJobPlacementShaper,
}
