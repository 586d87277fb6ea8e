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
from typing import Dict, List, Set

import numpy as np

from ..graphs import FWD, CompGraph, backward_name, partitioned_name

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
                         topology,
                         collective_time_cache: Dict = None) -> None:
    """Set every dep's init run time on the partitioned job given placement.

    Reference ``actions/utils.py:13-41`` + grouping + collective/one-to-one
    pricing.  ``op_name_to_worker`` maps partitioned-op NAME -> worker id.
    Vectorised over edges; collective times are memoised by their geometry
    key (message size, nodes, racks, cgs) in ``collective_time_cache`` (valid
    per-topology — all other model parameters are topology constants).
    """
    pg: CompGraph = partitioned_job.graph
    og: CompGraph = original_job.graph
    n_fwd = int((og.pass_type == FWD).sum())

    pg_idx = pg.name_to_idx
    rev_edge = pg.reverse_edge  # cached on the graph (shared across jobs)

    # dense per-node server index + coordinates
    w2n = topology.worker_to_node
    node_server = np.fromiter((w2n[op_name_to_worker[nm]] for nm in pg.names),
                              dtype=np.int64, count=pg.n)
    coords_arr = getattr(topology, "_coords_arr", None)
    if coords_arr is None:
        coords_arr = np.asarray(topology.coords, dtype=np.int64)
        topology._coords_arr = coords_arr
    node_coord = coords_arr[node_server]

    collectives: List[np.ndarray] = []
    one_to_one: List[np.ndarray] = []
    pair_lo_all: List[np.ndarray] = []
    pair_hi_all: List[np.ndarray] = []
    # per-class membership masks: the reference dedups via sets (an edge can
    # be gathered twice, e.g. the join edges appear in BOTH the fwd out-group
    # and the bwd in-group of the last forward op)
    coll_mask = np.zeros(pg.m, dtype=bool)
    o2o_mask = np.zeros(pg.m, dtype=bool)

    for f in range(og.n):
        if og.pass_type[f] != FWD:
            continue
        fwd_nm = og.names[f]
        bwd_nm = backward_name(fwd_nm, n_fwd)
        if fwd_nm in split_fwd_names:
            n_splits = fwd_name_to_splits[fwd_nm]
            fwd_deps = np.concatenate(
                [pg.out_edges_of(pg_idx[partitioned_name(fwd_nm, s)])
                 for s in range(n_splits)])
            in_e = np.concatenate(
                [pg.in_edges_of(pg_idx[partitioned_name(bwd_nm, s)])
                 for s in range(n_splits)])
            rev = rev_edge[in_e]
            sync_m = rev >= 0  # bidirectional sync edges
            bwd_deps = in_e[~sync_m]
            sync_e, sync_r = in_e[sync_m], rev[sync_m]
            # collective type 1: symmetric parent/child server multisets
            for ga in (fwd_deps, bwd_deps):
                parents = np.sort(node_server[pg.src[ga]])
                children = np.sort(node_server[pg.dst[ga]])
                if len(parents) == len(children) and np.array_equal(parents, children):
                    collectives.append(ga)
                    coll_mask[ga] = True
                else:
                    one_to_one.append(ga)
                    o2o_mask[ga] = True
            # collective type 2: each bidirectional sync pair is a collective
            if len(sync_e) > 0:
                lo = np.minimum(sync_e, sync_r)
                hi = np.maximum(sync_e, sync_r)
                _, first = np.unique(lo * pg.m + hi, return_index=True)
                first = np.sort(first)
                pair_lo_all.append(sync_e[first])
                pair_hi_all.append(sync_r[first])
                coll_mask[sync_e[first]] = True
                coll_mask[sync_r[first]] = True
        else:
            pf = pg_idx[str(fwd_nm)]
            pb = pg_idx[str(bwd_nm)]
            grp = np.unique(np.concatenate([pg.out_edges_of(pf),
                                            pg.in_edges_of(pb)]))
            one_to_one.append(grp)
            o2o_mask[grp] = True

    # NB collective multiset-equality note: node_server (dense server index)
    # stands in for the reference's worker-id strings — bijective with one
    # worker per server, so sorted-equality is unchanged.
    cache = collective_time_cache if collective_time_cache is not None else {}

    n_coll_deps = int(coll_mask.sum())
    n_o2o_deps = int(o2o_mask.sum())
    if pg.m != n_coll_deps + n_o2o_deps or (coll_mask & o2o_mask).any():
        raise AssertionError(
            f"partitioned graph has {pg.m} edges but grouped "
            f"{n_coll_deps} collective + {n_o2o_deps} one-to-one")

    # ---- price sync-pair collectives, batched ----
    if pair_lo_all:
        lo = np.concatenate(pair_lo_all)
        hi = np.concatenate(pair_hi_all)
        a, b = pg.src[lo], pg.dst[lo]  # the two endpoints of each pair
        msg = pg.size[lo] + pg.size[hi]
        same = node_server[a] == node_server[b]
        dn = (node_coord[a, 2] != node_coord[b, 2]).astype(np.int64)
        dr = (node_coord[a, 1] != node_coord[b, 1]).astype(np.int64)
        dc = (node_coord[a, 0] != node_coord[b, 0]).astype(np.int64)
        times = np.zeros(len(lo))
        active = ~same
        if active.any():
            combo = (dn + 2 * dr + 4 * dc)
            # few distinct (msg, geometry) combos per job: price each once
            keys = np.stack([msg, combo.astype(np.float64)], axis=1)
            uniq, inv = np.unique(keys[active], axis=0, return_inverse=True)
            priced = np.empty(len(uniq))
            for k, (m_, cb_) in enumerate(uniq):
                cb_ = int(cb_)
                key = (float(m_), 1 + (cb_ & 1), 1 + ((cb_ >> 1) & 1),
                       1 + ((cb_ >> 2) & 1))
                t = cache.get(key)
                if t is None:
                    t = calc_ramp_all_reduce_time(
                        message_size=key[0], node_ids=key[1], racks=key[2],
                        cgs=key[3], cont_racks=1,
                        x=topology.num_communication_groups,
                        DATA_RATE=topology.channel_bandwidth,
                        latency=topology.intra_gpu_propagation_latency,
                        IO_latency=topology.worker_io_latency)
                    cache[key] = t
                priced[k] = t
            times[active] = priced[inv]
        partitioned_job.dep_init_run_time[lo] = times
        partitioned_job.dep_init_run_time[hi] = times
        partitioned_job.dep_remaining[lo] = times
        partitioned_job.dep_remaining[hi] = times

    # ---- price group collectives (set_collective_dep_run_time:126-147) ----
    for ga in collectives:
        message_size = float(pg.size[ga].sum())
        endpoints = np.concatenate([pg.src[ga], pg.dst[ga]])
        coords = node_coord[endpoints]
        servers = node_server[endpoints]
        if len(np.unique(servers)) == 1:
            t = 0.0
        else:
            key = (message_size, len(np.unique(coords[:, 2])),
                   len(np.unique(coords[:, 1])), len(np.unique(coords[:, 0])))
            t = cache.get(key)
            if t is None:
                # cont_racks: the reference's "NEW NEW" dedup over (cg, rack,
                # server) tuples always yields 1 because the server id
                # determines its cg (actions/utils.py:221-232)
                t = calc_ramp_all_reduce_time(
                    message_size=message_size,
                    node_ids=key[1], racks=key[2], cgs=key[3],
                    cont_racks=1,
                    x=topology.num_communication_groups,
                    DATA_RATE=topology.channel_bandwidth,
                    latency=topology.intra_gpu_propagation_latency,
                    IO_latency=topology.worker_io_latency)
                cache[key] = t
        partitioned_job.dep_init_run_time[ga] = t
        partitioned_job.dep_remaining[ga] = t

    # ---- price one-to-one deps (set_one_to_one_dep_run_time:149-167) ----
    if one_to_one:
        o2o = np.concatenate(one_to_one)
        same_worker = node_server[pg.src[o2o]] == node_server[pg.dst[o2o]]
        # NB with one worker per server, same worker <=> same server
        sizes = pg.size[o2o]
        base = (topology.intra_gpu_propagation_latency
                + 2 * topology.worker_io_latency)
        times = np.where(same_worker | (sizes == 0), 0.0,
                         base + sizes / topology.channel_bandwidth)
        if np.isinf(times).any():
            raise ValueError("infinite one-to-one run time")
        partitioned_job.dep_init_run_time[o2o] = times
        partitioned_job.dep_remaining[o2o] = times
