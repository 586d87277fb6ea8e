"""RAMP-symmetry block-placement machinery.

Reference: ``ddls/environments/ramp_cluster/agents/placers/utils.py`` —
``dummy_ramp:235``, ``get_factor_pairs:445``, ``get_block_shapes:491``,
``get_block:464``, ``get_meta_block:193``, ``check_block:215``, ``ff_block:394``,
``ff_meta_block:133``, ``find_sub_block:385``, ``parent_collective_placement:258``,
``regular_collective_placement:333``, ``allocate:532``.
"""
from __future__ import annotations

import math
from typing import Dict, List, Sequence, Tuple

from ..graphs import CompGraph, FWD, backward_name, partitioned_name

Coord = Tuple[int, int, int]


def dummy_ramp(shape: Coord, cluster) -> Dict[Coord, dict]:
    """Dict mirror of the cluster: per-server free memory + job occupancy."""
    c, r, s = shape
    ramp = {}
    topo = cluster.topology
    for i in range(c):
        for j in range(r):
            for k in range(s):
                node = topo.name_to_node[f"{i}-{j}-{k}"]
                entry = {"mem": 0.0, "ops": [], "job_idxs": set()}
                for worker in topo.node_workers[node].values():
                    entry["mem"] += worker.memory_capacity - worker.memory_occupied
                    if len(worker.mounted_job_idx_to_ops) != 0:
                        entry["job_idxs"] = set(worker.mounted_job_idx_to_ops.keys())
                ramp[(i, j, k)] = entry
    return ramp


def get_factor_pairs(n: int) -> List[Tuple[int, int]]:
    return [(n // i, i) for i in range(1, n + 1) if n % i == 0]


def get_block_shapes(pairs: Sequence[Tuple[int, int]],
                     meta_block_shape: Coord) -> List[Coord]:
    blocks = []
    for p0, p1 in pairs:
        var = math.sqrt(p0)
        if (var % 1 == 0) and (var <= meta_block_shape[0]
                               and var <= meta_block_shape[1]
                               and p1 <= meta_block_shape[2]):
            blocks.append((int(var), int(var), p1))
        if p0 > meta_block_shape[0] or p0 > meta_block_shape[1] or p1 > meta_block_shape[2]:
            continue
        blocks.append((p0, 1, p1))
        blocks.append((p0, p1, 1))
    return blocks


def get_block(C: int, R: int, S: int, ramp_shape: Coord,
              origin: Coord = (0, 0, 0)) -> List[Coord]:
    block = []
    i, j, k = origin
    if S == -1:
        for n in range(C):
            block.append(((i + n) % (ramp_shape[0] + 1),
                          (j + n) % (ramp_shape[1] + 1),
                          k % ramp_shape[2]))
    else:
        for c in range(C):
            for r in range(R):
                for s in range(S):
                    block.append(((i + c) % ramp_shape[0],
                                  (j + r) % ramp_shape[1],
                                  (k + s) % ramp_shape[2]))
    return block


def get_meta_block(C: int, R: int, S: int, ramp_shape: Coord,
                   origin: Coord = (0, 0, 0)) -> List[Coord]:
    block = []
    i, j, k = origin
    for c in range(C):
        for r in range(R):
            for s in range(S):
                block.append(((i + c) % ramp_shape[0],
                              (j + r) % ramp_shape[1],
                              (k + s) % ramp_shape[2]))
    return block


def check_block(ramp: Dict[Coord, dict], block: List[Coord], op_size,
                job_idx) -> bool:
    """One-job-per-server + memory feasibility."""
    if block == []:
        return False
    for server in block:
        entry = ramp.get(server)
        if entry is None:
            return False
        if len(entry["job_idxs"]) != 0 and job_idx not in entry["job_idxs"]:
            return False
        if op_size is not None and entry["mem"] < op_size:
            return False
    return True


def ff_meta_block(block_shapes, ramp_shape, ramp, mode,
                  op_size=None, meta_block_origin: Coord = (0, 0, 0)):
    orgn_c, orgn_r, orgn_s = meta_block_origin
    for shape in block_shapes:
        I = ramp_shape[0] - shape[0] + 1
        J = ramp_shape[1] - shape[1] + 1
        K = ramp_shape[2] - shape[2] + 1
        if I <= 0 or J <= 0 or K <= 0:
            continue
        C, R, S = shape
        for i in range(ramp_shape[0]):
            for j in range(ramp_shape[1]):
                for k in range(ramp_shape[2]):
                    block = get_meta_block(C, R, S, ramp_shape,
                                           origin=(orgn_c + i, orgn_r + j, orgn_s + k))
                    if check_block(ramp, block, op_size, mode):
                        if mode == "sub":
                            return block
                        if mode == "meta":
                            return (block, shape, (orgn_c + i, orgn_r + j, orgn_s + k))
    return None


def find_meta_block(ramp, ramp_shape, meta_block_shape):
    return ff_meta_block([meta_block_shape], ramp_shape, ramp, "meta")


def ff_block(block_shapes, meta_shape, ramp_shape, ramp, mode, job_idx,
             op_size=None, meta_block_origin: Coord = (0, 0, 0)):
    orgn_c, orgn_r, orgn_s = meta_block_origin
    for shape in block_shapes:
        I = (meta_shape[0] - shape[0]) + 1
        J = (meta_shape[1] - shape[1]) + 1
        K = (meta_shape[2] - shape[2]) + 1
        if I <= 0 or J <= 0 or K <= 0:
            continue
        C, R, S = shape
        for i in range(I):
            for j in range(J):
                for k in range(K):
                    block = get_block(C, R, S, ramp_shape,
                                      origin=(orgn_c + i, orgn_r + j, orgn_s + k))
                    if check_block(ramp, block, op_size, job_idx):
                        return block
    return None


def find_sub_block(ramp, ramp_shape, meta_block_shape, meta_block_origin,
                   num_servers, op_size, job_idx):
    pairs = get_factor_pairs(num_servers)
    block_shapes = get_block_shapes(pairs, meta_block_shape)
    block_shapes += [(num_servers, num_servers, -1), (num_servers, 1, 1)]
    return ff_block(block_shapes, meta_block_shape, ramp_shape, ramp, "sub",
                    job_idx, op_size=op_size)


def check_meta_block_valid(c, r, s, ramp_topology, ramp_shape,
                           job_max_partition_degree, num_available_workers) -> bool:
    """Reference ``placers/utils.py:13-30``."""
    if job_max_partition_degree <= c * r * s <= min(num_available_workers,
                                                    job_max_partition_degree):
        if c * r * s == job_max_partition_degree:
            if c == r and find_meta_block(ramp_topology, ramp_shape, (c, r, s)) is not None:
                return True
        else:
            if find_meta_block(ramp_topology, ramp_shape, (c, r, s)) is not None:
                return True
    return False


# ---------------------------------------------------------------------------
# per-op allocation over the forward graph
# ---------------------------------------------------------------------------

def get_allocation_preamble(forward_graph: CompGraph, mp_split_names, mp_splits):
    """Topo sequence + per-op split counts + parents map
    (reference ``placers/utils.py:68-98``)."""
    order = forward_graph.topo_order_fwd_subgraph()
    sequence = [forward_graph.names[i] for i in order]
    parents = {}
    for i in range(forward_graph.n):
        if forward_graph.pass_type[i] != FWD:
            continue
        ps = []
        for e in forward_graph.in_edges_of(i):
            u = int(forward_graph.src[e])
            if forward_graph.pass_type[u] == FWD:
                ps.append(forward_graph.names[u])
        parents[forward_graph.names[i]] = ps
    split_list = list(mp_split_names)
    splits = []
    for s in sequence:
        if s in split_list:
            splits.append(mp_splits[split_list.index(s)])
        else:
            splits.append(1)
    op_server_info = {forward_graph.names[i]: []
                      for i in range(forward_graph.n)
                      if forward_graph.pass_type[i] == FWD}
    return sequence, splits, op_server_info, parents


def parent_collective_placement(ramp, fwd_mem: Dict[str, float], num_fwd: int,
                                op: str, split: int, meta_block_info, parents,
                                op_server_info):
    """Co-locate split children on (one of) their parents' server sets
    (reference ``placers/utils.py:258-314``)."""
    op_requirement = fwd_mem[op]
    backward_op_id = backward_name(op, num_fwd)
    meta_block = meta_block_info[0]
    meta_set = set(meta_block)
    parents_servers = []
    for parent in parents[op]:
        if set(op_server_info[parent]).issubset(meta_set):
            parents_servers.append(op_server_info[parent])
    for servers in parents_servers:
        if split != len(servers):
            continue
        available = sum(ramp[server]["mem"] for server in servers)
        if available >= op_requirement:
            i = 0
            while i < split:
                for server in servers:
                    ramp[server]["mem"] -= op_requirement / split
                    if split > 1:
                        ramp[server]["ops"].append(partitioned_name(op, i))
                        ramp[server]["ops"].append(partitioned_name(backward_op_id, i))
                    else:
                        ramp[server]["ops"].append(op)
                        ramp[server]["ops"].append(backward_op_id)
                    op_server_info[op].append(server)
                    i += 1
            return ramp, op_server_info
    return None


def regular_collective_placement(ramp, ramp_shape, fwd_mem: Dict[str, float],
                                 num_fwd: int, op: str, split: int,
                                 meta_block_info, op_server_info, job_idx):
    """First-fit a split op onto a symmetric sub-block
    (reference ``placers/utils.py:333-383``)."""
    meta_block, meta_block_shape, meta_block_origin = meta_block_info
    backward_op_id = backward_name(op, num_fwd)
    num_servers = split
    if num_servers > len(meta_block):
        return None
    op_size = fwd_mem[op] / split
    block = find_sub_block(ramp, ramp_shape, meta_block_shape, meta_block_origin,
                           num_servers, op_size, job_idx)
    if not block:
        return None
    for j in range(len(block)):
        ramp[block[j]]["mem"] -= op_size
        if split > 1:
            ramp[block[j]]["ops"].append(partitioned_name(op, j))
            ramp[block[j]]["ops"].append(partitioned_name(backward_op_id, j))
        else:
            ramp[block[j]]["ops"].append(op)
            ramp[block[j]]["ops"].append(backward_op_id)
        op_server_info[op].append(block[j])
    return ramp, op_server_info


def allocate(ramp, ramp_shape, fwd_mem: Dict[str, float], num_fwd: int,
             sequence, splits, meta_block_info, parents, op_server_info, job_idx):
    """Allocate every (split) op: parent-collective first, then regular
    (reference ``placers/utils.py:532-582``)."""
    for i in range(len(sequence)):
        op, split = sequence[i], splits[i]
        alloc = parent_collective_placement(ramp, fwd_mem, num_fwd, op, split,
                                            meta_block_info, parents, op_server_info)
        if not alloc:
            alloc = regular_collective_placement(ramp, ramp_shape, fwd_mem, num_fwd,
                                                 op, split, meta_block_info,
                                                 op_server_info, job_idx)
        if not alloc:
            return None
        ramp, op_server_info = alloc
    return ramp, op_server_info
