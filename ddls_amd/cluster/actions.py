"""Cluster control-plane actions.

Reference: ``ddls/environments/ramp_cluster/actions/`` — ``op_partition.py:8``,
``op_placement.py:7``, ``op_schedule.py:3``, ``dep_placement.py:6``,
``dep_schedule.py:3``, ``action.py:3``.

Conventions in this rebuild: op identity inside an action is the partitioned-op
NAME (str); dep identity is the dense edge index into the partitioned job's
CompGraph; worker/channel ids are strings.
"""
from __future__ import annotations

import copy
from collections import defaultdict
from typing import Dict, Optional, Set

from ..jobs import Job
from .comm_model import update_dep_run_times
from .partition import build_partitioned_graph


class OpPartition:
    """job_id -> op_name -> num_partitions; builds partitioned Job objects.

    Partition degrees must be even (or 1).  Reuses the cluster's
    (model, max-degree) memo of partitioned graphs + immutable details
    (reference ``op_partition.py:26-64``).
    """

    def __init__(self, action: Dict, cluster):
        # normalise op keys to str names
        self.action = {jid: {str(op): int(n) for op, n in ops.items()}
                       for jid, ops in action.items()}

        self.job_id_to_mp_split_forward_op_ids = defaultdict(list)
        self.job_id_to_mp_splits = defaultdict(list)
        self.job_id_to_forward_op_id_to_mp_splits = defaultdict(dict)
        self.job_id_to_max_partition_degree = defaultdict(lambda: 1)
        for job_id, ops in self.action.items():
            for op_name, num in ops.items():
                if num != 1 and num % 2 != 0:
                    raise ValueError(
                        f"num_partitions={num} for job {job_id} op {op_name}: "
                        "RAMP expects even partition degrees")
                if num > 1:
                    self.job_id_to_mp_split_forward_op_ids[job_id].append(op_name)
                    self.job_id_to_mp_splits[job_id].append(num)
                    self.job_id_to_forward_op_id_to_mp_splits[job_id][op_name] = num
                    if num > self.job_id_to_max_partition_degree[job_id]:
                        self.job_id_to_max_partition_degree[job_id] = num

        self.job_ids: Set = set()
        self.partitioned_jobs: Dict[int, Job] = {}
        self.original_jobs: Dict[int, Job] = {}
        for job_id in self.action:
            job = cluster.job_queue.jobs[job_id]
            self.job_ids.add(job_id)
            self.original_jobs[job_id] = job

            model = job.details["model"]
            degree = self.job_id_to_max_partition_degree[job_id]
            memo = cluster.job_model_to_max_num_partitions_to_init_details
            entry = memo[model].get(degree)
            if entry is None or entry.get("graph") is None:
                pgraph = build_partitioned_graph(
                    job.graph,
                    mp_split_ids=self.job_id_to_mp_split_forward_op_ids[job_id],
                    mp_splits=self.job_id_to_mp_splits[job_id],
                    model=model)
                memo[model][degree] = entry = {"graph": pgraph, "immutable": None}
            pgraph = entry["graph"]

            # structured shallow copy == deepcopy for this dict's shapes
            # (scalars, flat dicts of scalars, sets of ids, read-only arrays);
            # generic deepcopy recursion was ~0.2 ms per env step.  The
            # reference's overlay quirk (original-job detail values win over
            # the partitioned job's recomputed ones, job.py:364-378) is
            # preserved because every key is still carried over.
            details = {}
            for k, v in job.details.items():
                if isinstance(v, set):
                    details[k] = set(v)
                elif isinstance(v, dict):
                    details[k] = dict(v)
                elif isinstance(v, list):
                    details[k] = list(v)
                else:
                    details[k] = v  # scalars / read-only numpy arrays
            details["max_partitions_per_op"] = degree
            self.partitioned_jobs[job_id] = Job(
                graph=pgraph,
                num_training_steps=job.num_training_steps,
                max_acceptable_job_completion_time_frac=job.max_acceptable_job_completion_time_frac,
                job_id=job_id,
                original_job=job,
                details=details,
                immutable=entry["immutable"])
            if entry["immutable"] is None:
                entry["immutable"] = self.partitioned_jobs[job_id].immutable

    def __len__(self):
        return len(self.action)


class OpPlacement:
    """job_id -> op_name -> worker_id.  Constructing it prices every dep of the
    partitioned jobs (collective grouping + analytic model), like the reference
    ctor (``op_placement.py:28-30``)."""

    def __init__(self, action: Dict, op_partition: OpPartition, cluster):
        self.action = action
        self.job_ids, self.worker_ids = set(), set()
        self.worker_to_ops = defaultdict(list)
        self.job_id_to_worker_ids = defaultdict(set)
        for job_id, ops in action.items():
            self.job_ids.add(job_id)
            for op_name, worker_id in ops.items():
                self.worker_ids.add(worker_id)
                self.worker_to_ops[worker_id].append(
                    {"op_id": op_name, "job_id": job_id})
                self.job_id_to_worker_ids[job_id].add(worker_id)

        for job_id in self.job_ids:
            partitioned = op_partition.partitioned_jobs[job_id]
            original = op_partition.original_jobs[job_id]
            update_dep_run_times(
                partitioned_job=partitioned,
                original_job=original,
                split_fwd_names=set(op_partition.job_id_to_mp_split_forward_op_ids[job_id]),
                op_name_to_worker=action[job_id],
                fwd_name_to_splits=op_partition.job_id_to_forward_op_id_to_mp_splits[job_id],
                topology=cluster.topology,
                collective_time_cache=getattr(cluster, "collective_time_cache", None))


class OpSchedule:
    """worker_id -> job_id -> op_name -> priority (higher value preferred)."""

    def __init__(self, action: Dict):
        self.action = action
        self.job_ids = set()
        for worker_id in self.action:
            for job_id in self.action[worker_id]:
                self.job_ids.add(job_id)


class DepSchedule:
    """channel_id -> job_id -> dep_idx -> priority."""

    def __init__(self, action: Dict):
        self.action = action
        self.job_ids = set()
        for channel_id in self.action:
            for job_id in self.action[channel_id]:
                self.job_ids.add(job_id)


class DepPlacement:
    """job_id -> dep_idx -> set of channel_ids ({None} for non-flows)."""

    def __init__(self, action: Dict):
        self.action = action
        self.job_ids = set(action.keys())
        self.channel_ids = set()
        self.job_to_dep_to_channels = action  # alias: same mapping shape
        for job_id, deps in action.items():
            # NB the reference keeps channel None entries for non-flow deps
            # (dep_placement.py:19-34); the None "channel" carries the job
            # through DepSchedule into the Action intersection
            for channels in deps.values():
                self.channel_ids.update(channels)


class JobPlacementShape:
    """job_id -> (c, r, s) meta-block shape for the shaping env
    (reference ``actions/job_placement_shape.py:1-21``)."""

    def __init__(self, action: Dict):
        self.action = dict(action)
        self.job_ids = set(self.action.keys())

    def __len__(self):
        return len(self.action)


class Action:
    """Bundle of the five sub-actions; a job survives only if EVERY sub-action
    handled it (set intersection, reference ``action.py:36-52``).  An optional
    ``job_placement_shape`` joins the intersection (used by the shaping env)."""

    KEYS = ("op_partition", "op_placement", "op_schedule", "dep_placement",
            "dep_schedule")

    def __init__(self,
                 op_partition: Optional[OpPartition] = None,
                 op_placement: Optional[OpPlacement] = None,
                 op_schedule: Optional[OpSchedule] = None,
                 dep_placement: Optional[DepPlacement] = None,
                 dep_schedule: Optional[DepSchedule] = None,
                 job_placement_shape: Optional[JobPlacementShape] = None):
        self.actions = defaultdict(lambda: None)
        for key, act in zip(self.KEYS, (op_partition, op_placement, op_schedule,
                                        dep_placement, dep_schedule)):
            if act is not None:
                self.actions[key] = act

        self.cause_of_unsuccessful_handling = None
        if len(self.actions) > 0:
            self.job_ids = set.intersection(
                *[set(a.job_ids) for a in self.actions.values()])
            if job_placement_shape is not None:
                self.job_ids &= set(job_placement_shape.job_ids)
            self.job_idxs = set(
                op_partition.partitioned_jobs[jid].details["job_idx"]
                for jid in self.job_ids)
            for key, act in self.actions.items():
                if len(act.action) == 0:
                    self.cause_of_unsuccessful_handling = key
                    break
        else:
            self.job_ids = set()
            self.job_idxs = set()

        for key, act in self.actions.items():
            if act.job_ids == self.job_ids:
                continue  # every handled job survived: nothing to filter
            self._filter_action(key, act)

    def _filter_action(self, key, act):
        if key in {"op_partition", "op_placement", "dep_placement"}:
            for jid in list(act.action.keys()):
                if jid not in self.job_ids:
                    del act.action[jid]
        elif key in {"op_schedule", "dep_schedule"}:
            for device_id in act.action:
                for jid in list(act.action[device_id].keys()):
                    if jid not in self.job_ids:
                        del act.action[device_id][jid]
        else:
            raise ValueError(f"unrecognised action key {key}")
