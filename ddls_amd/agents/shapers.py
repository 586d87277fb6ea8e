"""Job placement shapers: choose a valid (c, r, s) meta-block shape per job.

Reference: ``agents/job_placement_shapers/ramp_random_job_placement_shaper.py:10``,
``ramp_first_fit_job_placement_shaper.py:10`` and
``placers/utils.py:32-65`` (get_partitioned_job_valid_meta_block_shapes).
"""
from __future__ import annotations

import numpy as np

from ..cluster.actions import JobPlacementShape, OpPartition
from .placement_utils import check_meta_block_valid, dummy_ramp


def get_partitioned_job_valid_meta_block_shapes(cluster,
                                                job_max_partition_degree: int):
    """(action_set, action_mask) over all (c, r, s) shapes."""
    ramp_shape = cluster.topology.shape
    ramp_topology = dummy_ramp(ramp_shape, cluster)
    num_avail = cluster.topology.num_workers - len(cluster.mounted_workers)
    action_set, action_mask = [], []
    for c in range(1, ramp_shape[0] + 1):
        for r in range(1, ramp_shape[1] + 1):
            for s in range(1, ramp_shape[2] + 1):
                action_set.append((c, r, s))
                action_mask.append(check_meta_block_valid(
                    c, r, s, ramp_topology, ramp_shape,
                    job_max_partition_degree, num_avail))
    return np.array(action_set), np.array(action_mask, dtype=bool)


class RampFirstFitJobPlacementShaper:
    def get(self, op_partition: OpPartition, cluster) -> JobPlacementShape:
        job_to_shape = {}
        for job_id in op_partition.partitioned_jobs:
            shapes, mask = get_partitioned_job_valid_meta_block_shapes(
                cluster, op_partition.job_id_to_max_partition_degree[job_id])
            valid = shapes[mask]
            if len(valid) > 0:
                job_to_shape[job_id] = tuple(int(x) for x in valid[0])
        return JobPlacementShape(job_to_shape)


class RampRandomJobPlacementShaper:
    def get(self, op_partition: OpPartition, cluster) -> JobPlacementShape:
        job_to_shape = {}
        for job_id in op_partition.partitioned_jobs:
            shapes, mask = get_partitioned_job_valid_meta_block_shapes(
                cluster, op_partition.job_id_to_max_partition_degree[job_id])
            valid = shapes[mask]
            if len(valid) > 0:
                idx = np.random.randint(len(valid))
                job_to_shape[job_id] = tuple(int(x) for x in valid[idx])
        return JobPlacementShape(job_to_shape)
