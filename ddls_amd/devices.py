"""Worker processors and link channels.

Reference: ``ddls/devices/processors/processor.py:3``, ``gpus/A100.py:7``,
``ddls/devices/channels/channel.py:7``.  The native worker model here is the
MI355X (288 GB HBM3E); an A100 model is kept for parity runs against the
reference's profiled workloads.
"""
from __future__ import annotations

from collections import defaultdict
from typing import Union


def gen_channel_id(src, dst, channel_number) -> str:
    return f"{src}|{dst}|{channel_number}"


class Processor:
    device_type: str = "abstract"
    memory_capacity: int = 0

    def __init__(self, processor_id=None):
        self.processor_id = processor_id if processor_id is not None else id(self)
        self.reset()

    def __str__(self):
        return f"{self.device_type}_{self.processor_id}"

    def reset(self):
        self.memory_occupied = 0.0
        self.mounted_job_idx_to_ops = defaultdict(set)
        self.mounted_job_op_to_priority = {}
        self.mounted_job_idx_to_job_id = {}

    def mount(self, job, op_idx: int):
        """Mount op (dense index into job.graph) onto this worker.

        Mirrors ``A100.mount`` (``A100.py:38-49``): memory accounting + per-job
        op sets; raises on over-commit or missing device profile.
        """
        g = job.graph
        if self.device_type not in g.compute_cost:
            raise KeyError(
                f"op profiled for {list(g.compute_cost)} but worker is {self.device_type}")
        mem = float(g.memory_cost[op_idx])
        if self.memory_occupied + mem > self.memory_capacity:
            raise MemoryError(
                f"allocating {mem} B for job {job.job_id} op {g.names[op_idx]} "
                f"but only {self.memory_capacity - self.memory_occupied} B free on "
                f"{self.processor_id}")
        job_idx = job.details["job_idx"]
        self.mounted_job_idx_to_ops[job_idx].add(op_idx)
        self.mounted_job_idx_to_job_id[job_idx] = job.job_id
        self.memory_occupied += mem

    def unmount(self, job, op_idx: int):
        job_idx = job.details["job_idx"]
        self.memory_occupied -= float(job.graph.memory_cost[op_idx])
        self.mounted_job_idx_to_ops[job_idx].discard(op_idx)
        self.mounted_job_op_to_priority.pop((job_idx, op_idx), None)
        if len(self.mounted_job_idx_to_ops[job_idx]) == 0:
            del self.mounted_job_idx_to_ops[job_idx]
            self.mounted_job_idx_to_job_id.pop(job_idx, None)


class MI355X(Processor):
    """AMD Instinct MI355X: 288 GB HBM3E (native worker model of this rebuild)."""
    device_type = "MI355X"
    memory_capacity = int(288e9)


class A100(Processor):
    """Reference parity worker (``A100.py:15-17``): 80 GB."""
    device_type = "A100"
    memory_capacity = int(80e9)


class GPU(Processor):
    """Generic parametric GPU (reference ``gpus/gpu.py:5-51``): memory /
    compute-unit count / clock are constructor parameters, so experiments can
    model arbitrary accelerators without subclassing.  ``device_type`` keys the
    per-device ``compute_cost`` dicts, as for the named processors."""
    device_type = "GPU"

    def __init__(self, processor_id=None, memory_capacity: int = int(80e9),
                 num_compute_units: int = 256, clock_frequency: float = 2.4e9,
                 device_type: str = "GPU"):
        self.memory_capacity = int(memory_capacity)
        self.num_compute_units = num_compute_units
        self.clock_frequency = clock_frequency
        self.device_type = device_type
        super().__init__(processor_id)


class Channel:
    """Directed per-link channel (reference ``channel.py:7-42``)."""

    def __init__(self, src, dst, channel_number: int,
                 channel_bandwidth: Union[int, float] = int(1.25e9)):
        self.src = src
        self.dst = dst
        self.channel_number = channel_number if channel_number is not None else id(self)
        self.channel_id = gen_channel_id(src, dst, self.channel_number)
        self.channel_bandwidth = channel_bandwidth
        self.reset()

    def __str__(self):
        return f"Channel_{self.channel_id}"

    def reset(self):
        # job_idx -> number of mounted deps (the sim only ever needs the
        # one-job-per-channel occupancy test; per-dep scheduling priorities
        # live on the job's dense arrays)
        self.mounted_job_idx_to_deps = {}

    def mount(self, job, dep_idx: int, count: int = 1):
        job_idx = job.details["job_idx"]
        self.mounted_job_idx_to_deps[job_idx] = (
            self.mounted_job_idx_to_deps.get(job_idx, 0) + count)

    def unmount(self, job, dep_idx: int, count: int = 1):
        job_idx = job.details["job_idx"]
        remaining = self.mounted_job_idx_to_deps.get(job_idx, 0) - count
        if remaining <= 0:
            self.mounted_job_idx_to_deps.pop(job_idx, None)
        else:
            self.mounted_job_idx_to_deps[job_idx] = remaining
