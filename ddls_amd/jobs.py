"""Jobs: computation-graph workloads with a tick state machine.

Reference: ``ddls/demands/jobs/job.py:42`` (Job), ``jobs_generator.py:64``
(JobsGenerator), ``ddls/environments/cluster/job_queue.py:8`` (JobQueue).

The rebuild separates the immutable graph structure (:class:`CompGraph`, shared
between all jobs of one model+partitioning) from the per-job mutable tick state
(numpy arrays), so memoised graph reuse never aliases running-job state.
"""
from __future__ import annotations

import copy
import glob as globlib
from collections import OrderedDict, defaultdict
from typing import Dict, List, Optional, Union

import numpy as np

from .distributions import Distribution, ListOfDistributions, distribution_from_config
from .graphs import CompGraph, load_pipedream_graph
from .utils import Sampler


class GraphImmutableDetails:
    """Per-(model, partition-degree) immutable job details
    (reference ``job.py:192-212`` _init_job_immutable_details)."""

    __slots__ = ("max_compute_node", "max_compute_cost", "max_memory_node",
                 "max_memory_cost", "max_throughput_node", "max_node_throughput",
                 "max_depth_node", "max_depth", "node_to_depth",
                 "max_dep_size_dep", "max_dep_size",
                 "job_sequential_completion_time_per_step",
                 "job_total_op_memory_cost", "job_total_dep_size")

    def __deepcopy__(self, memo):
        return self  # immutable

    def __init__(self, graph: CompGraph):
        self.max_compute_node, self.max_compute_cost = {}, {}
        self.max_throughput_node, self.max_node_throughput = {}, {}
        for dt, cc in graph.compute_cost.items():
            i = int(np.argmax(cc))
            self.max_compute_node[dt] = i
            self.max_compute_cost[dt] = float(cc[i])
            with np.errstate(divide="ignore", invalid="ignore"):
                thr = np.where(cc > 0, graph.memory_cost / np.where(cc > 0, cc, 1), 0.0)
            j = int(np.argmax(thr))
            self.max_throughput_node[dt] = j
            self.max_node_throughput[dt] = float(thr[j])
        mi = int(np.argmax(graph.memory_cost))
        self.max_memory_node = mi
        self.max_memory_cost = float(graph.memory_cost[mi])
        self.node_to_depth = graph.node_depths()
        di = int(np.argmax(self.node_to_depth))
        self.max_depth_node = di
        self.max_depth = int(self.node_to_depth[di])
        if graph.m > 0:
            ei = int(np.argmax(graph.size))
            self.max_dep_size_dep = ei
            self.max_dep_size = float(graph.size[ei])
        else:
            self.max_dep_size_dep, self.max_dep_size = None, 0.0
        self.job_sequential_completion_time_per_step = {
            dt: float(cc.sum()) for dt, cc in graph.compute_cost.items()}
        self.job_total_op_memory_cost = graph.total_memory_cost()
        self.job_total_dep_size = graph.total_dep_size()


class Job:
    """A DNN training job: run its computation graph ``num_training_steps`` times."""

    def __init__(self,
                 graph: CompGraph,
                 num_training_steps: int,
                 max_acceptable_job_completion_time_frac: float,
                 job_id: Optional[int] = None,
                 original_job: Optional["Job"] = None,
                 details: Optional[dict] = None,
                 immutable: Optional[GraphImmutableDetails] = None):
        if not (0 < max_acceptable_job_completion_time_frac <= 1):
            raise ValueError(
                "max_acceptable_job_completion_time_frac must be in (0, 1] but is "
                f"{max_acceptable_job_completion_time_frac}")
        self.graph = graph
        self.num_training_steps = num_training_steps
        self.max_acceptable_job_completion_time_frac = max_acceptable_job_completion_time_frac
        self.training_step_counter = 0
        self._job_id = job_id if job_id is not None else id(self)
        self.details = dict(details) if details else {}
        self.reset_job(self.details, immutable=immutable)
        self.original_job = original_job if original_job is not None else self

    # runtime details rebuilt by reset_job — must NOT be shared with a clone
    _RUNTIME_DETAIL_KEYS = frozenset((
        "mounted_workers", "mounted_channels", "communication_overhead_time",
        "computation_overhead_time", "max_acceptable_job_completion_time",
        "job_sequential_completion_time", "job_total_op_memory_cost",
        "job_total_dep_size", "max_compute_node", "max_compute_cost",
        "max_memory_node", "max_memory_cost", "max_depth", "node_to_depth",
        "max_node_throughput", "max_dep_size_dep", "max_dep_size"))

    def clone(self) -> "Job":
        """Fast copy of a PRISTINE (never-run) job — what ``Sampler`` needs
        per sample.  Shares the immutable graph + GraphImmutableDetails and
        rebuilds the tick state fresh; generic ``copy.deepcopy`` recursion
        over the details dict was ~0.5 ms per env step."""
        import copy as _copy
        overlay = {}
        for k, v in self.details.items():
            if k in self._RUNTIME_DETAIL_KEYS:
                continue  # rebuilt by reset_job from the immutable details
            if isinstance(v, (set, dict, list)):
                overlay[k] = _copy.deepcopy(v)
            else:
                overlay[k] = v
        return Job(graph=self.graph,
                   num_training_steps=self.num_training_steps,
                   max_acceptable_job_completion_time_frac=
                   self.max_acceptable_job_completion_time_frac,
                   job_id=self._job_id, details=overlay,
                   immutable=self.immutable)

    # ---- identity ----
    @property
    def job_id(self):
        return self._job_id

    @job_id.setter
    def job_id(self, value):
        if self.original_job is not self:
            self.original_job.job_id = value
        self._job_id = value

    @property
    def model(self) -> str:
        return self.details.get("model", self.graph.model)

    # ---- reset ----
    def reset_job(self, details: Optional[dict] = None,
                  graph: Optional[CompGraph] = None,
                  immutable: Optional[GraphImmutableDetails] = None):
        """Reference ``job.py:327-385`` reset_job."""
        if graph is not None:
            self.graph = graph
        self.immutable = immutable if immutable is not None else GraphImmutableDetails(self.graph)
        im = self.immutable
        self.job_total_operation_memory_cost = im.job_total_op_memory_cost
        self.job_total_dependency_size = im.job_total_dep_size

        self.reset_job_training_step(preserve_mounts=hasattr(self, "mounted_device_type"))

        seq = {dt: v * self.num_training_steps
               for dt, v in im.job_sequential_completion_time_per_step.items()}
        self.details.update({
            "model": self.details.get("model", self.graph.model),
            "job_sequential_completion_time": seq,
            "job_total_op_memory_cost": im.job_total_op_memory_cost,
            "job_total_dep_size": im.job_total_dep_size,
            "max_compute_node": im.max_compute_node,
            "max_compute_cost": im.max_compute_cost,
            "max_memory_node": im.max_memory_node,
            "max_memory_cost": im.max_memory_cost,
            "max_depth": im.max_depth,
            "node_to_depth": im.node_to_depth,
            "max_node_throughput": im.max_node_throughput,
            "max_dep_size_dep": im.max_dep_size_dep,
            "max_dep_size": im.max_dep_size,
        })
        # snapshot the caller's overlay BEFORE resetting mutable details, then
        # fresh-init mutable details and overlay on top: reference order
        # (job.py:364-378: immutable -> mutable -> max_acceptable -> update(details))
        overlay = dict(details) if details else {}
        overlay.pop("job_sequential_completion_time", None)
        self.details["communication_overhead_time"] = 0.0
        self.details["computation_overhead_time"] = 0.0
        self.details["mounted_workers"] = set()
        self.details["mounted_channels"] = set()
        self.details["max_acceptable_job_completion_time"] = {
            dt: self.max_acceptable_job_completion_time_frac * v for dt, v in seq.items()}
        self.details.update(overlay)

    def reset_job_training_step(self, preserve_mounts: bool = False):
        """Reset the per-training-step tick state (reference ``job.py:387-484``)."""
        g = self.graph
        n, m = g.n, g.m
        if not preserve_mounts:
            self.mounted_device_type: List[Optional[str]] = [None] * n
        self.op_remaining = np.full(n, np.nan)
        for i, dt in enumerate(self.mounted_device_type):
            if dt is not None:
                self.op_remaining[i] = g.compute_cost[dt][i]
        self.dep_init_run_time = np.full(m, np.nan)
        self.dep_remaining = np.full(m, np.nan)

        self.ops_ready = np.zeros(n, dtype=bool)
        self.ops_completed = np.zeros(n, dtype=bool)
        self.deps_ready = np.zeros(m, dtype=bool)
        self.deps_completed = np.zeros(m, dtype=bool)
        self.parent_deps_completed = np.zeros(n, dtype=np.int64)
        self.ops_ready[g.source_nodes()] = True
        self.num_ops_completed = 0
        self.num_deps_completed = 0

    # ---- lifecycle registration (reference job.py:394-430) ----
    def register_job_arrived(self, time_arrived, job_idx: int):
        self.details["time_arrived"] = time_arrived
        self.details["time_started"] = None
        self.details["time_completed"] = None
        self.details["job_idx"] = job_idx
        self.original_job.details["job_idx"] = job_idx

    def register_job_running(self, time_started):
        self.details["time_started"] = time_started

    def register_job_completed(self, time_completed):
        self.details["time_completed"] = time_completed

    # ---- per-op / per-dep state ----
    def reset_op_remaining_run_time(self, op_idx: int, device_type: Optional[str]):
        if device_type is not None:
            self.op_remaining[op_idx] = self.graph.compute_cost[device_type][op_idx]
        else:
            self.op_remaining[op_idx] = np.nan
        self.mounted_device_type[op_idx] = device_type

    def set_dep_init_run_time(self, dep_idx: int, run_time):
        self.dep_init_run_time[dep_idx] = np.nan if run_time is None else run_time
        self.dep_remaining[dep_idx] = self.dep_init_run_time[dep_idx]

    def reset_dep_remaining_run_time(self, dep_idx: int):
        self.dep_remaining[dep_idx] = self.dep_init_run_time[dep_idx]

    # ---- tick state machine (reference job.py:486-563) ----
    def tick_op(self, op_idx: int, tick: float):
        rem = self.op_remaining[op_idx]
        rem -= min(tick, rem)
        self.op_remaining[op_idx] = rem
        if rem == 0:
            self._register_completed_op(op_idx)

    def _register_completed_op(self, op_idx: int):
        self.ops_completed[op_idx] = True
        self.ops_ready[op_idx] = False
        self.num_ops_completed += 1
        for e in self.graph.out_edges_of(op_idx):
            self.deps_ready[e] = True
        if self.is_training_step_complete():
            self.training_step_counter += 1

    def tick_dep(self, dep_idx: int, tick: float):
        rem = self.dep_remaining[dep_idx]
        rem -= min(tick, rem)
        self.dep_remaining[dep_idx] = rem
        if rem == 0:
            self._register_completed_dep(dep_idx)

    def _register_completed_dep(self, dep_idx: int):
        if self.deps_completed[dep_idx]:
            return
        self.deps_completed[dep_idx] = True
        self.deps_ready[dep_idx] = False
        self.num_deps_completed += 1
        child = int(self.graph.dst[dep_idx])
        self.parent_deps_completed[child] += 1
        # NB reference quirk (job.py:531): readiness fires when the number of
        # completed in-deps (sync in-edges included) EQUALS the number of
        # non-bidirectional parents.
        if self.parent_deps_completed[child] == self.graph.true_parent_count[child]:
            self.ops_ready[child] = True

    def is_training_step_complete(self) -> bool:
        return (self.num_ops_completed == self.graph.n
                and self.num_deps_completed == self.graph.m)

    def is_job_complete(self) -> bool:
        return self.training_step_counter == self.num_training_steps

    def __str__(self):
        return (f"Job(id={self.job_id}, model={self.model}, n={self.graph.n}, "
                f"m={self.graph.m}, steps={self.num_training_steps})")


class JobQueue:
    """FIFO queue with capacity (reference ``job_queue.py:8-39``)."""

    def __init__(self, queue_capacity: int = 10):
        self.queue_capacity = queue_capacity
        self.jobs: "OrderedDict[int, Job]" = OrderedDict()

    def can_fit(self, job: Job) -> bool:
        return len(self.jobs) + 1 <= self.queue_capacity

    def add(self, job: Job):
        if not self.can_fit(job):
            raise RuntimeError("job queue full")
        self.jobs[job.job_id] = job

    def remove(self, job: Job):
        self.jobs.pop(job.job_id, None)

    def __len__(self):
        return len(self.jobs)


class JobsGenerator:
    """Loads/replicates job graph profiles and samples jobs + interarrival times.

    Reference ``jobs_generator.py:64-333``.  Accepts either ``path_to_files``
    (pipedream ``.txt`` profiles) or ``graphs`` (pre-built CompGraphs, used by
    tests and the synthetic workload generator).
    """

    def __init__(self,
                 path_to_files: Optional[str] = None,
                 job_interarrival_time_dist: Union[Distribution, dict, None] = None,
                 max_acceptable_job_completion_time_frac_dist=None,
                 max_files: Optional[int] = None,
                 replication_factor: int = 1,
                 job_sampling_mode: str = "remove_and_repeat",
                 shuffle_files: bool = False,
                 num_training_steps: int = 1,
                 max_partitions_per_op_in_observation: int = 1,
                 graphs: Optional[List[CompGraph]] = None,
                 processor_type_profiled: str = "A100"):
        if graphs is None:
            if path_to_files is None:
                raise ValueError("need path_to_files or graphs")
            files = sorted(globlib.glob(path_to_files.rstrip("/") + "/*"))
            files = [f for f in files if f.split(".")[-1] in ("txt", "pbtxt")]
            if max_files is not None:
                files = files[:max_files]
            if not files:
                raise FileNotFoundError(f"no job profile files in {path_to_files}")
            from .graphs import load_pbtxt_graph
            graphs = [load_pbtxt_graph(f, processor_type_profiled)
                      if f.endswith(".pbtxt")
                      else load_pipedream_graph(f, processor_type_profiled)
                      for f in files]

        self.job_interarrival_time_dist = distribution_from_config(
            job_interarrival_time_dist if job_interarrival_time_dist is not None
            else {"_target_": "ddls_amd.distributions.Fixed", "val": 1000})

        frac_dist = max_acceptable_job_completion_time_frac_dist
        if frac_dist is None:
            frac_dist = {"_target_": "ddls_amd.distributions.Fixed", "val": 1.0}
        frac_dist = distribution_from_config(frac_dist) if isinstance(frac_dist, dict) else frac_dist
        if isinstance(frac_dist, ListOfDistributions):
            frac_dist = frac_dist.sample()
        self.max_acceptable_job_completion_time_frac_dist = frac_dist

        # per-model immutable details memo (reference jobs_generator.py:140-146)
        self.model_to_immutable: Dict[str, GraphImmutableDetails] = {}
        jobs: List[Job] = []
        i = 0
        for _ in range(replication_factor):
            for g in graphs:
                model = g.model
                if model not in self.model_to_immutable:
                    self.model_to_immutable[model] = GraphImmutableDetails(g)
                job = Job(graph=g,
                          num_training_steps=num_training_steps,
                          max_acceptable_job_completion_time_frac=float(
                              self.max_acceptable_job_completion_time_frac_dist.sample()),
                          job_id=i,
                          details={"model": model},
                          immutable=self.model_to_immutable[model])
                jobs.append(job)
                i += 1

        self.job_sampler = Sampler(pool=jobs, sampling_mode=job_sampling_mode,
                                   shuffle=shuffle_files)
        self.max_partitions_per_op_in_observation = max_partitions_per_op_in_observation
        self.jobs_params = self._init_jobs_params(jobs, max_partitions_per_op_in_observation)

    def __len__(self):
        return len(self.job_sampler)

    def reset(self, resample_fracs: bool = True):
        """Reset for a new episode without rebuilding the pool (the reference
        constructs a fresh JobsGenerator per reset; this is the O(1)
        equivalent: refill the sampler and resample each prototype's
        max-acceptable-JCT fraction)."""
        self.job_sampler.reset()
        if resample_fracs:
            for job in self.job_sampler.original_pool:
                frac = float(self.max_acceptable_job_completion_time_frac_dist.sample())
                job.max_acceptable_job_completion_time_frac = frac
                seq = job.details["job_sequential_completion_time"]
                job.details["max_acceptable_job_completion_time"] = {
                    dt: frac * v for dt, v in seq.items()}
            self.jobs_params = self._init_jobs_params(
                self.job_sampler.original_pool,
                self.max_partitions_per_op_in_observation)

    def sample_job(self) -> Job:
        return self.job_sampler.sample()

    def sample_interarrival_time(self, size=None):
        if len(self.job_sampler) == 0:
            return float("inf")
        return self.job_interarrival_time_dist.sample(size=size)

    def _init_jobs_params(self, jobs: List[Job], max_partitions: int) -> dict:
        """Normalisation stats for observations (reference
        ``jobs_generator.py:276-333``, incl. the fully-connected max-dep-size
        assumption)."""
        params = defaultdict(list)
        device_type = list(jobs[0].details["job_sequential_completion_time"].keys())[0]
        for job in jobs:
            params["job_sequential_completion_times"].append(
                job.details["job_sequential_completion_time"][device_type])
            params["max_acceptable_job_completion_times"].append(
                job.details["max_acceptable_job_completion_time"][device_type])
            params["max_acceptable_job_completion_time_fracs"].append(
                job.max_acceptable_job_completion_time_frac)
            params["job_total_op_memory_costs"].append(job.details["job_total_op_memory_cost"])
            params["job_total_dep_sizes"].append(job.details["job_total_dep_size"])
            params["job_total_num_ops"].append(job.graph.n)
            params["job_total_num_deps"].append(job.graph.m)
            params["job_num_training_steps"].append(job.num_training_steps)
            params["job_max_op_compute_throughputs"].append(
                job.details["max_node_throughput"][device_type])
            params["job_max_dep_size"].append(job.details["max_dep_size"])

        out = {}
        for key, vals in params.items():
            out[key] = vals
            out[f"min_{key}"] = float(np.min(vals))
            if key == "job_total_num_ops":
                out[f"max_{key}"] = int(np.max(vals) * max_partitions)
            elif key == "job_total_num_deps":
                max_forward_edges = int((np.max(vals) / 2) * max_partitions * 2)
                out[f"max_{key}"] = max_forward_edges + int(max_forward_edges * 2)
            elif key == "job_total_dep_sizes":
                max_nodes = np.max(params["job_total_num_ops"]) * max_partitions
                fully_connected = int(max_nodes * (max_nodes - 1) / 2)
                out[f"max_{key}"] = float(np.max(vals)) * fully_connected
            else:
                out[f"max_{key}"] = float(np.max(vals))
        return out
