"""Analytic communication-time model + collective grouping.

Reference: ``ddls/environments/ramp_cluster/actions/utils.py`` —
``calc_ramp_all_reduce_collective_communication_run_time:42``,
``calc_one_to_one_communication_run_time:90``, ``effective_trx_per_comm:101``,
``get_parallel_add_comp_time:108``, ``group_deps_into_collective_and_one_to_one_
communications:247``, ``set_collective_dep_run_time:126``.

The model is kept parameterised so the same machinery prices both the RAMP
optical fabric (per-transceiver DATA_RATE = total/num_comm_groups) and an
MI355X xGMI node profile (7 point-to-point links x ~153 GB/s per GPU — see
``xgmi_profile`` below): ring all-reduce on xGMI is per-link bound, so the
effective bandwidth term is per-link, not switched.
"""
from __future__ import annotations

import math
from typing import Dict, List, Sequence, Set, Tuple

import numpy as np

from ..graphs import BWD, FWD, CompGraph, backward_name, partitioned_name

# device compute-side constants of the reference model (actions/utils.py:52-60)
DEFAULT_MEM_FRQ = 2e12
DEFAULT_PI = 130e12
DEFAULT_BYTES_PER_COMP = 2


def effective_trx_per_comm(cg: int = 32, d: int = 32, J: int = 1) -> float:
    if d == 1:
        return 0
    spare = min(cg // J, cg // (d - 1)) - 1
    return 1 + spare


def get_parallel_add_comp_time(data_sz: float, devices: int,
                               MEM_FRQ: float = DEFAULT_MEM_FRQ,
                               pi: float = DEFAULT_PI,
                               bytes_per_comp: float = DEFAULT_BYTES_PER_COMP) -> float:
    n_op = math.ceil(math.log2(devices)) if devices > 1 else 0
    n_bytes = (devices + 1) * bytes_per_comp
    ai = n_op / n_bytes
    total_ops = n_op * (data_sz / devices) / bytes_per_comp
    return total_ops / min(MEM_FRQ * ai, pi)


def calc_ramp_all_reduce_time(message_size: float,
                              node_ids: int, racks: int, cgs: int,
                              cont_racks: int = 1,
                              x: int = 32,
                              DATA_RATE: float = 1.6e12,
                              MEM_FRQ: float = DEFAULT_MEM_FRQ,
                              latency: float = 1.25e-6,
                              pi: float = DEFAULT_PI,
                              bytes_per_comp: float = DEFAULT_BYTES_PER_COMP,
                              IO_latency: float = 100e-9) -> float:
    """4 hierarchical subgroup steps; x2 for reduce-scatter + all-gather."""
    data_per_tx = DATA_RATE / x
    subgroup_size = [cgs, min(cgs, node_ids), racks, math.ceil(node_ids / x)]
    effect_bw = [effective_trx_per_comm(cg=x, d=d, J=cont_racks) * data_per_tx
                 for d in subgroup_size]
    msg_size = [math.ceil(message_size / subgroup_size[0])]
    for s in subgroup_size[1:]:
        msg_size.append(math.ceil(msg_size[-1] / s))
    comm_time = 0.0
    comp_time = 0.0
    for step, sub in enumerate(subgroup_size):
        if sub > 1:
            comp_time += get_parallel_add_comp_time(
                msg_size[step] * sub, devices=sub, MEM_FRQ=MEM_FRQ, pi=pi,
                bytes_per_comp=bytes_per_comp)
            comm_time += latency + 2 * IO_latency + msg_size[step] / effect_bw[step]
    total = 2 * comm_time + comp_time
    if math.isinf(total):
        raise ValueError("infinite ramp all-reduce collective run time")
    return total


def calc_one_to_one_time(message_size: float,
                         DATA_RATE: float = 1.6e12,
                         latency: float = 1.25e-6,
                         IO_latency: float = 100e-9) -> float:
    t = latency + 2 * IO_latency + message_size / DATA_RATE
    if math.isinf(t):
        raise ValueError("infinite one-to-one run time")
    return t


def xgmi_profile() -> dict:
    """MI355X 8-GPU node as a RAMP-style parameter set: 7 xGMI p2p links per
    GPU at ~153 GB/s each (~1.07 TB/s aggregate); ~1 us software latency.
    Use via topology kwargs for MI355X-flavoured simulations."""
    return {
        "num_communication_groups": 7,
        "total_node_bandwidth": int(7 * 153e9),
        "intra_gpu_propagation_latency": 1e-6,
        "worker_io_latency": 100e-9,
    }


def update_dep_run_times(partitioned_job,
                         original_job,
                         split_fwd_names: Set[str],
                         op_name_to_worker: Dict[str, str],
                         fwd_name_to_splits: Dict[str, int],
                         topology) -> None:
    """Set every dep's init run time on the partitioned job given placement.

    Reference ``actions/utils.py:13-41`` + grouping + collective/one-to-one
    pricing.  ``op_name_to_worker`` maps partitioned-op NAME -> worker id.
    """
    pg: CompGraph = partitioned_job.graph
    og: CompGraph = original_job.graph
    n_fwd = int((og.pass_type == FWD).sum())

    pg_idx = pg.name_to_idx
    # edge lookup (u_idx, v_idx) -> edge index
    edge_lookup = {(int(u), int(v)): e
                   for e, (u, v) in enumerate(zip(pg.src, pg.dst))}

    worker_of = {}  # pg node idx -> worker id

    def worker(node_idx: int) -> str:
        w = worker_of.get(node_idx)
        if w is None:
            w = op_name_to_worker[pg.names[node_idx]]
            worker_of[node_idx] = w
        return w

    collectives: List[List[int]] = []
    one_to_one: Set[int] = set()
    collective_deps: Set[int] = set()

    for f in range(og.n):
        if og.pass_type[f] != FWD:
            continue
        fwd_nm = og.names[f]
        bwd_nm = backward_name(fwd_nm, n_fwd)
        if fwd_nm in split_fwd_names:
            n_splits = fwd_name_to_splits[fwd_nm]
            fwd_deps: List[int] = []
            bwd_deps: List[int] = []
            sync_deps: List[int] = []
            sync_pairs_added = set()
            for s in range(n_splits):
                pf = pg_idx[partitioned_name(fwd_nm, s)]
                fwd_deps.extend(int(e) for e in pg.out_edges_of(pf))
                pb = pg_idx[partitioned_name(bwd_nm, s)]
                for e in pg.in_edges_of(pb):
                    e = int(e)
                    u, v = int(pg.src[e]), int(pg.dst[e])
                    if (v, u) in edge_lookup:  # bidirectional sync edge
                        key = (min(u, v), max(u, v))
                        if key not in sync_pairs_added:
                            sync_pairs_added.add(key)
                            sync_deps.append(e)
                            sync_deps.append(edge_lookup[(v, u)])
                    else:
                        bwd_deps.append(e)
            # collective type 1: symmetric parent/child server multisets
            for group in (fwd_deps, bwd_deps):
                parents = sorted(worker(int(pg.src[e])) for e in group)
                children = sorted(worker(int(pg.dst[e])) for e in group)
                if parents == children:
                    collectives.append(list(group))
                    collective_deps.update(group)
                else:
                    one_to_one.update(group)
            # collective type 2: each bidirectional sync pair is a collective
            for i in range(0, len(sync_deps), 2):
                pair = [sync_deps[i], sync_deps[i + 1]]
                collectives.append(pair)
                collective_deps.update(pair)
        else:
            pf = pg_idx[str(fwd_nm)]
            one_to_one.update(int(e) for e in pg.out_edges_of(pf))
            pb = pg_idx[str(bwd_nm)]
            one_to_one.update(int(e) for e in pg.in_edges_of(pb))

    if pg.m != len(collective_deps) + len(one_to_one):
        raise AssertionError(
            f"partitioned graph has {pg.m} edges but grouped "
            f"{len(collective_deps)} collective + {len(one_to_one)} one-to-one")

    # ---- price collectives (set_collective_dep_run_time:126-147) ----
    for group in collectives:
        cgs, racks, nodes, servers = set(), set(), set(), set()
        message_size = 0.0
        for e in group:
            for node_idx in (int(pg.src[e]), int(pg.dst[e])):
                w = worker(node_idx)
                server_node = topology.worker_to_node[w]
                c, r, s = topology.coords[server_node]
                cgs.add(c)
                racks.add(r)
                nodes.add(s)
                servers.add(w)
            message_size += float(pg.size[e])
        if len(servers) == 1:
            t = 0.0
        else:
            # cont_racks: the reference's "NEW NEW" dedup over (cg, rack,
            # server) tuples always yields 1 because the server id determines
            # its cg (actions/utils.py:221-232)
            t = calc_ramp_all_reduce_time(
                message_size=message_size,
                node_ids=len(nodes), racks=len(racks), cgs=len(cgs),
                cont_racks=1,
                x=topology.num_communication_groups,
                DATA_RATE=topology.channel_bandwidth,
                latency=topology.intra_gpu_propagation_latency,
                IO_latency=topology.worker_io_latency)
        for e in group:
            partitioned_job.set_dep_init_run_time(e, t)

    # ---- price one-to-one deps (set_one_to_one_dep_run_time:149-167) ----
    for e in one_to_one:
        u, v = int(pg.src[e]), int(pg.dst[e])
        if worker(u) == worker(v):
            t = 0.0
        elif pg.size[e] == 0:
            t = 0.0
        else:
            t = calc_one_to_one_time(
                float(pg.size[e]),
                DATA_RATE=topology.channel_bandwidth,
                latency=topology.intra_gpu_propagation_latency,
                IO_latency=topology.worker_io_latency)
        partitioned_job.set_dep_init_run_time(e, t)
