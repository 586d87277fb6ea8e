"""Op placers and dep (flow) placer.

Reference: ``agents/placers/ramp_first_fit_op_placer.py:23``,
``random_op_placer.py:13``, ``first_fit_dep_placer.py:18``.
"""
from __future__ import annotations

import random
from collections import defaultdict
from typing import Dict

import numpy as np

from ..cluster.actions import DepPlacement, OpPartition, OpPlacement
from ..devices import gen_channel_id
from ..graphs import FWD, backward_name, partitioned_name
from .placement_utils import allocate, dummy_ramp, get_allocation_preamble


class RampFirstFitOpPlacer:
    """Topo-sort the forward graph and allocate each (split) op to symmetric
    server blocks, treating the whole cluster as the meta-block
    (reference ``ramp_first_fit_op_placer.py:23-115``)."""

    def __init__(self, **kwargs):
        pass

    def get(self, op_partition: OpPartition, cluster, verbose: bool = False,
            job_placement_shape=None) -> OpPlacement:
        from .placement_utils import find_meta_block
        ramp_shape = cluster.topology.shape
        ramp_topology = dummy_ramp(ramp_shape, cluster)

        job_to_op_to_worker: Dict = {}
        for job_id in op_partition.action:
            partitioned_job = op_partition.partitioned_jobs[job_id]
            job_idx = partitioned_job.details["job_idx"]
            original_job = cluster.job_queue.jobs[job_id]
            og = original_job.graph
            num_fwd = int((og.pass_type == FWD).sum())
            fwd_mem = {og.names[i]: float(og.memory_cost[i])
                       for i in range(og.n) if og.pass_type[i] == FWD}

            mp_split_names = op_partition.job_id_to_mp_split_forward_op_ids[job_id]
            mp_splits = op_partition.job_id_to_mp_splits[job_id]
            sequence, splits, op_server_info, parents = get_allocation_preamble(
                og, mp_split_names, mp_splits)

            if job_placement_shape is not None:
                # shaping env: pack the job into a meta-block of the chosen
                # shape (reference shaping-variant placer)
                shape = job_placement_shape.action.get(job_id)
                if shape is None:
                    continue
                meta_block_info = find_meta_block(ramp_topology, ramp_shape,
                                                  tuple(shape))
                if meta_block_info is None:
                    continue
            else:
                # PAC-ML env: whole cluster as meta-block (reference :78-82)
                servers = [cluster.topology.coords[i]
                           for i in range(cluster.topology.num_nodes)]
                meta_block_info = (servers, ramp_shape, (0, 0, 0))

            allocated = allocate(ramp_topology, ramp_shape, fwd_mem, num_fwd,
                                 sequence, splits, meta_block_info, parents,
                                 op_server_info, job_idx)
            if not allocated:
                continue
            ramp_topology, op_server_info = allocated

            placement = {}
            op_to_split = dict(zip(sequence, splits))
            for op, servers_of_op in op_server_info.items():
                split = op_to_split[op]
                bwd = backward_name(op, num_fwd)
                for i, server in enumerate(servers_of_op):
                    node = cluster.topology.coord_to_node[server]
                    # one worker per server (RAMP restriction)
                    worker_id = next(iter(cluster.topology.node_workers[node].keys()))
                    if split > 1:
                        placement[partitioned_name(op, i)] = worker_id
                        placement[partitioned_name(bwd, i)] = worker_id
                    else:
                        placement[op] = worker_id
                        placement[bwd] = worker_id
            job_to_op_to_worker[job_id] = placement

        return OpPlacement(job_to_op_to_worker, op_partition=op_partition,
                           cluster=cluster)


class RandomOpPlacer:
    """Random memory-feasible worker per op, respecting the one-job-per-worker
    rule (reference ``random_op_placer.py:13-95``)."""

    def __init__(self, **kwargs):
        pass

    def get(self, op_partition: OpPartition, cluster, verbose: bool = False) -> OpPlacement:
        job_to_op_to_worker: Dict = {}
        # free memory + occupancy snapshot
        free_mem = {w.processor_id: w.memory_capacity - w.memory_occupied
                    for w in cluster.workers}
        occupancy = {w.processor_id: set(w.mounted_job_idx_to_ops.keys())
                     for w in cluster.workers}
        for job_id in op_partition.action:
            job = op_partition.partitioned_jobs[job_id]
            job_idx = job.details["job_idx"]
            g = job.graph
            placement = {}
            ok = True
            for i in range(g.n):
                candidates = [wid for wid in free_mem
                              if free_mem[wid] >= g.memory_cost[i]
                              and (not occupancy[wid] or job_idx in occupancy[wid])]
                if not candidates:
                    ok = False
                    break
                wid = random.choice(candidates)
                placement[g.names[i]] = wid
                free_mem[wid] -= g.memory_cost[i]
                occupancy[wid].add(job_idx)
            if ok:
                job_to_op_to_worker[job_id] = placement
        return OpPlacement(job_to_op_to_worker, op_partition=op_partition,
                           cluster=cluster)


class FirstFitDepPlacer:
    """First fit over shortest paths x shuffled channel numbers; the whole job
    is dropped if any flow is unplaceable (reference
    ``first_fit_dep_placer.py:18-170``)."""

    def __init__(self, **kwargs):
        pass

    def get(self, op_partition: OpPartition, op_placement: OpPlacement, cluster,
            verbose: bool = False) -> DepPlacement:
        new_placements = op_placement.action
        job_to_dep_to_channels: Dict = defaultdict(lambda: defaultdict(set))
        if len(new_placements) == 0:
            return DepPlacement(dict(job_to_dep_to_channels))

        topo = cluster.topology
        channels_used_across_jobs = set()
        for job_id, job in op_partition.partitioned_jobs.items():
            if job_id not in new_placements:
                continue
            used_for_this_job = set()
            g = job.graph
            placement = new_placements[job_id]
            job_idx = job.details["job_idx"]
            # vectorised flow classification
            node_of = np.array([topo.worker_to_node[placement[nm]]
                                for nm in g.names], dtype=np.int64)
            src_nodes = node_of[g.src]
            dst_nodes = node_of[g.dst]
            is_flow = (src_nodes != dst_nodes) & (g.size > 0)
            dropped = False
            # same job may reuse a (src,dst) channel freely; cache only with a
            # single channel per link (multi-channel keeps the reference's
            # per-flow shuffled channel choice)
            cache_pairs = topo.num_channels == 1
            pair_cache = {}
            for e in range(g.m):
                if is_flow[e]:
                    pair = (int(src_nodes[e]), int(dst_nodes[e]))
                    found = pair_cache.get(pair) if cache_pairs else None
                    if found is None:
                        found = self._find_path_channel(
                            cluster, pair[0], pair[1], job_idx,
                            channels_used_across_jobs)
                        if cache_pairs:
                            pair_cache[pair] = found
                    if found is None:
                        job_to_dep_to_channels.pop(job_id, None)
                        dropped = True
                        break
                    path, channel_num = found
                    direct = getattr(topo, "direct_cid", None)
                    if direct is not None and len(path) == 2:
                        cid = direct[(path[0], path[1], channel_num)]
                        job_to_dep_to_channels[job_id][e] = (cid,)
                        used_for_this_job.add(cid)
                    else:
                        cids = set()
                        for idx in range(len(path) - 1):
                            sN, d = path[idx], path[idx + 1]
                            cid = gen_channel_id(topo.node_names[sN],
                                                 topo.node_names[d], channel_num)
                            cids.add(cid)
                            used_for_this_job.add(cid)
                        job_to_dep_to_channels[job_id][e] = cids
                else:
                    job_to_dep_to_channels[job_id][e] = (None,)
            # reference tracks channels used even by dropped jobs
            channels_used_across_jobs |= used_for_this_job

        return DepPlacement({k: dict(v) for k, v in job_to_dep_to_channels.items()})

    def _find_path_channel(self, cluster, src_node, dst_node, job_idx,
                           channels_used_across_jobs):
        topo = cluster.topology
        direct = getattr(topo, "direct_cid", None)
        if direct is not None:  # full mesh: single-hop paths
            if topo.num_channels == 1:
                channel_nums = (0,)
            else:
                channel_nums = list(range(topo.num_channels))
                random.shuffle(channel_nums)
            for channel_num in channel_nums:
                cid = direct[(src_node, dst_node, channel_num)]
                channel = topo.channel_id_to_channel[cid]
                if (job_idx in channel.mounted_job_idx_to_deps
                        or (len(channel.mounted_job_idx_to_deps) == 0
                            and cid not in channels_used_across_jobs)):
                    return [src_node, dst_node], channel_num
            return None
        paths = topo.shortest_paths(src_node, dst_node)
        channel_nums = list(range(topo.num_channels))
        random.shuffle(channel_nums)
        for path in paths:
            for channel_num in channel_nums:
                valid = True
                for idx in range(len(path) - 1):
                    s, d = path[idx], path[idx + 1]
                    cid = gen_channel_id(topo.node_names[s], topo.node_names[d],
                                         channel_num)
                    channel = topo.channel_id_to_channel[cid]
                    if job_idx not in channel.mounted_job_idx_to_deps:
                        if (len(channel.mounted_job_idx_to_deps) > 0
                                or cid in channels_used_across_jobs):
                            valid = False
                            break
                if valid:
                    return path, channel_num
        return None
