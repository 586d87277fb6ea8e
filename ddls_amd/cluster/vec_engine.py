"""Vectorised RAMP env engine: the steady-state PAC-ML env step over B envs
as compact SoA state, steppable either by the CPU mirror here or by ONE HIP
kernel launch (``ops/hip/env_step.hip``) — the GPU-resident rollout core of
this rebuild (reference hot loop
``ddls/environments/ramp_cluster/ramp_cluster_environment.py:379-467,562-651,894-1044``
and ``ramp_job_partitioning_environment.py:300-400``).

Key observation (the reference's own design pivot,
``ramp_cluster_environment.py:84-93`` + memo tables ``:269-277``): once the
(model, partition-degree) lookahead memo is warm, a RampJobPartitioning step
reduces to
  1. expand action -> per-op splits -> first-fit block placement SEARCH over
     the occupied server set (``agents/placers/utils.py:532``),
  2. a memo probe for the lookahead JCT (device-side open-addressing hash
     table here — SURVEY K4),
  3. accept/block against the job's max-acceptable JCT,
  4. the outer event loop (advance wall clock to completions/arrivals),
  5. observation encode (static per-model features + a few dynamic scalars)
     and the action mask.
Everything per-step therefore fits a compact per-env state (server occupancy
bits, running-job slots, arrival schedule cursor) — exactly what a batched
HIP kernel wants.  Job arrival schedules are action-independent, so they are
drained from the REAL JobsGenerator per (env, episode) on the host (RNG
sequence identical to the reference env) and uploaded as arrays.

Parity contract (tested): CPU mirror == RampJobPartitioningEnvironment
step-for-step (obs/reward/done, f64); HIP kernel == CPU mirror bitwise.
"""
from __future__ import annotations

import math
from dataclasses import dataclass, field
from typing import Dict, List, Optional, Tuple

import numpy as np

from ..agents.partitioners import sip_ml_num_partitions
from ..agents.placement_utils import (get_block, get_block_shapes,
                                      get_factor_pairs)
from ..graphs import FWD
from ..utils import seed_everything

# job-log / slot status codes
PENDING, RUNNING, COMPLETED, BLOCKED = 0, 1, 2, 3
# per-step env status codes (kernel <-> host protocol)
ST_IDLE, ST_STEP, ST_OK, ST_MISS, ST_ERR = 0, 1, 2, 3, 4

HASH_EMPTY = np.uint64(0xFFFFFFFFFFFFFFFF)


def _hash_key(model_id: int, degree: int) -> int:
    return (int(model_id) << 20) | int(degree)


def _hash_slot(key: int, hsize: int) -> int:
    # multiplicative hash; hsize power of two
    return (key * 0x9E3779B97F4A7C15 >> 40) & (hsize - 1)


# ---------------------------------------------------------------------------
# Spec compilation
# ---------------------------------------------------------------------------

@dataclass
class ModelSpec:
    name: str
    n: int                      # original graph nodes
    m: int                      # original graph edges
    seq_total: float            # sequential JCT x num_training_steps
    mem_total: float            # job_total_op_memory_cost (original)
    dep_total: float            # job_total_dep_size (original)
    gf_static: np.ndarray       # [17] f64; entries 3,4,5,15,16 overwritten/step
    node_features: np.ndarray   # [n, 5] f32 (static per model)
    edge_features: np.ndarray   # [m, 2] f32
    edges_src: np.ndarray       # [m] i64
    edges_dst: np.ndarray       # [m] i64
    action_to_degree: np.ndarray  # [A] i32 (0 for action 0)
    # placement program (original forward graph)
    seq_len: int = 0
    op_mem: np.ndarray = None     # [seq_len] f64 fwd memory cost
    par_ptr: np.ndarray = None    # [seq_len+1] i32
    par_idx: np.ndarray = None    # [] i32 parent positions in sequence
    num_training_steps: int = 1


@dataclass
class MdSpec:
    """Per-(model, degree) canonical entry (empty-cluster pipeline result;
    shape-independent quantities — see ``batched_lookahead`` caveat)."""
    model_id: int
    degree: int
    placeable: bool
    splits: np.ndarray          # [seq_len] i32
    n_workers: int = 0
    n_channels: int = 0
    flow_size: float = 0.0
    pj_n: int = 0
    pj_m: int = 0
    pj_mem: float = 0.0
    pj_dep: float = 0.0
    # partitioned job's sequential JCT (ulp-differs from the original's: the
    # split ops re-sum cc/k); used by speedup stats + reward normalisers
    pj_seq: float = 0.0


@dataclass
class RewardSpec:
    """LookaheadJobCompletionTime / JobAcceptance / MultiObjective params."""
    kind: str = "lookahead_job_completion_time"
    sign: float = -1.0
    inverse: bool = False
    transform_with_log: bool = False
    normaliser: int = 0         # 0 none, 1 seq_jct, 2 seq_jct*fail_factor
    fail_factor: float = 1.0
    fail_const: Optional[float] = None   # None -> job seq jct
    # acceptance component (JobAcceptance / MultiObjective)
    acc_success: float = 1.0
    acc_fail: float = -1.0
    jct_weight: float = 1.0
    blocking_weight: float = 0.0  # 0 -> pure JCT reward
    # throughput family (reference mean_compute_throughput.py:9-57 etc.):
    # 0 none, 1 compute (partitioned op memory), 2 cluster (op memory +
    # dep size), 3 demand_total (ORIGINAL job's totals).  Reward = the
    # first cluster step's info_processed / step_time (the RL env extracts
    # the reward right after cluster.step(action), before fast-forward).
    # CPU-mirror support; the GPU kernel port is pending (GpuEngine gates).
    tp_kind: int = 0


@dataclass
class EngineSpec:
    C: int
    R: int
    S: int
    W: int                      # num servers (== workers)
    A: int                      # action-space size (max_partitions+1)
    eps: float                  # machine epsilon (1e-7)
    max_sim: float
    mem_capacity: float
    infinite_pool: bool         # remove_and_repeat / replace sampling
    static_shape_ok: np.ndarray  # [A] u8 (mask feasibility, occupancy-free)
    models: List[ModelSpec] = field(default_factory=list)
    mds: List[MdSpec] = field(default_factory=list)
    md_index: Dict[Tuple[int, int], int] = field(default_factory=dict)
    # placement cache ((mdi, occ bytes) -> frozen union or None): the
    # search is deterministic in (md, occupancy), and occupancy patterns
    # repeat heavily in steady state (empty cluster being the common
    # case) — same insight as the env's empty-cluster pipeline cache
    # (ramp_job_partitioning.py); bounded, reset when full
    place_cache: Dict[tuple, Optional[tuple]] = field(default_factory=dict)
    # shape lists per split count k: (C,R,S) triples, S=-1 diagonal sentinel
    shape_ptr: np.ndarray = None   # [max_split+2] i32
    shapes: np.ndarray = None      # [n_shapes, 3] i32
    reward: RewardSpec = None
    max_running: int = 0
    device_type: str = "A100"
    # lookahead memo values (host mirror of the device hash table):
    # (model_id, degree) -> (jct_total, comm_oh_total, comp_oh_total,
    #                        active_time_sum_1step)
    memo: Dict[Tuple[int, int], Tuple[float, float, float, float]] = \
        field(default_factory=dict)

    def model_id(self, name: str) -> int:
        return self._name_to_id[name]


def _shape_lists(max_split: int, ramp_shape) -> Tuple[np.ndarray, np.ndarray]:
    """find_sub_block's candidate shape order per split count
    (``agents/placement_utils.py:152-158``)."""
    ptr = [0]
    out = []
    for k in range(0, max_split + 1):
        if k >= 1:
            shapes = get_block_shapes(get_factor_pairs(k), ramp_shape)
            shapes = shapes + [(k, k, -1), (k, 1, 1)]
            out.extend(shapes)
        ptr.append(len(out))
    return (np.asarray(ptr, dtype=np.int32),
            np.asarray(out, dtype=np.int32).reshape(-1, 3))


def _static_mask_ok(A: int, ramp_shape) -> np.ndarray:
    """Occupancy-free part of the action mask (reference
    ``ramp_job_partitioning_observation.py:80-131`` never checks occupancy
    beyond the worker COUNT)."""
    ok = np.zeros(A, dtype=np.uint8)
    ok[0] = 1
    for a in range(1, A):
        if not ((a > 1 and a % 2 == 0) or a == 1):
            continue
        if a == 1:
            ok[a] = 1
            continue
        pairs = get_factor_pairs(a)
        block_shapes = get_block_shapes(pairs, ramp_shape)
        b = []
        for shape in block_shapes:
            b.extend(get_block(shape[0], shape[1], shape[2], ramp_shape))
        ok[a] = 1 if len(b) > 0 else 0
    return ok


def _graph_features_static(job, cluster) -> np.ndarray:
    """f64 mirror of RampJobPartitioningObservation._graph_features for the
    episode-independent entries (3,4,5,15,16 are placeholders)."""
    g = job.graph
    d = job.details
    p = cluster.jobs_generator.jobs_params
    dt = next(iter(cluster.topology.worker_types))

    def norm(val, lo, hi):
        return (val - lo) / (hi - lo) if hi - lo != 0 else 1.0

    feats = [
        norm(g.n, p["min_job_total_num_ops"], p["max_job_total_num_ops"]),
        norm(g.m, p["min_job_total_num_deps"], p["max_job_total_num_deps"]),
        norm(d["job_sequential_completion_time"][dt],
             p["min_job_sequential_completion_times"],
             p["max_job_sequential_completion_times"]),
        0.0, 0.0, 0.0,   # 3,4: max-acceptable norms (per episode); 5: frac
        norm(d["job_total_op_memory_cost"],
             p["min_job_total_op_memory_costs"],
             p["max_job_total_op_memory_costs"]),
        norm(d["job_total_dep_size"], p["min_job_total_dep_sizes"],
             p["max_job_total_dep_sizes"]),
        norm(job.num_training_steps, p["min_job_num_training_steps"],
             p["max_job_num_training_steps"]),
    ]
    op_cc, op_mc = [], []
    for dtt in cluster.topology.worker_types:
        mx = d["max_compute_cost"][dtt]
        op_cc.extend((g.compute_cost[dtt] / mx if mx != 0 else
                      np.zeros(g.n)).tolist())
    op_mc = (g.memory_cost / d["max_memory_cost"]
             if d["max_memory_cost"] != 0 else np.zeros(g.n)).tolist()
    feats.append(float(np.mean(op_cc)))
    feats.append(float(np.median(op_cc)))
    feats.append(float(np.mean(op_mc)))
    feats.append(float(np.median(op_mc)))
    dep_sizes = g.size
    feats.append(float(np.mean(dep_sizes) / d["max_dep_size"])
                 if d["max_dep_size"] else 0.0)
    feats.append(float(np.median(dep_sizes) / d["max_dep_size"])
                 if d["max_dep_size"] else 0.0)
    feats.extend([0.0, 0.0])   # 15,16: dynamic cluster features
    return np.asarray(feats, dtype=np.float64)


def _reward_spec_from_env(env) -> RewardSpec:
    from ..envs.rewards import (JobAcceptance, LookaheadJobCompletionTime,
                                MeanClusterThroughput, MeanComputeThroughput,
                                MeanDemandTotalThroughput,
                                MultiObjectiveJCTBlocking)
    rf = env.reward_function
    norm_map = {None: 0, "job_sequential_completion_time": 1,
                "job_sequential_completion_time_times_fail_reward_factor": 2}

    def jct_part(r, out: RewardSpec):
        out.sign = float(r.sign)
        out.inverse = bool(r.inverse)
        out.transform_with_log = bool(r.transform_with_log)
        out.normaliser = norm_map[r.normaliser]
        out.fail_factor = float(r.fail_reward_factor)
        if isinstance(r.fail_reward, (int, float)):
            out.fail_const = float(r.fail_reward)
        elif r.fail_reward == "job_sequential_completion_time":
            out.fail_const = None
        else:
            raise ValueError(f"engine: unsupported fail_reward {r.fail_reward}")

    out = RewardSpec()
    if isinstance(rf, LookaheadJobCompletionTime):
        out.kind = "lookahead_job_completion_time"
        jct_part(rf, out)
        out.jct_weight, out.blocking_weight = 1.0, 0.0
    elif isinstance(rf, JobAcceptance):
        out.kind = "job_acceptance"
        out.acc_success = float(rf.success_reward)
        out.acc_fail = float(rf.fail_reward)
        out.jct_weight, out.blocking_weight = 0.0, 1.0
    elif isinstance(rf, MultiObjectiveJCTBlocking):
        out.kind = "multi_objective_jct_blocking"
        jct_part(rf.jct, out)
        out.acc_success = float(rf.acceptance.success_reward)
        out.acc_fail = float(rf.acceptance.fail_reward)
        out.jct_weight = float(rf.jct_weight)
        out.blocking_weight = float(rf.blocking_weight)
    elif isinstance(rf, MeanComputeThroughput):
        out.kind, out.tp_kind = "mean_compute_throughput", 1
        out.jct_weight = out.blocking_weight = 0.0
    elif isinstance(rf, MeanClusterThroughput):
        out.kind, out.tp_kind = "mean_cluster_throughput", 2
        out.jct_weight = out.blocking_weight = 0.0
    elif isinstance(rf, MeanDemandTotalThroughput):
        out.kind, out.tp_kind = "mean_demand_total_throughput", 3
        out.jct_weight = out.blocking_weight = 0.0
    else:
        raise ValueError(
            f"engine: unsupported reward function {type(rf).__name__}")
    return out


def compile_engine_spec(env, lookahead_device=None,
                        verbose: bool = False) -> EngineSpec:
    """Build the engine spec from a (reset) RampJobPartitioningEnvironment.

    Runs the real partition->placement pipeline per (model, degree) on the
    EMPTY cluster to extract canonical shape-independent quantities, and the
    (batched HIP when available) lookahead for the memo values — the same
    startup precompute as ``batched_lookahead.precompute_lookahead_memos``.
    """
    from .batched_lookahead import mount_job_for_lookahead, run_lookahead_batch
    from .actions import OpPartition

    cluster = env.cluster
    topo = cluster.topology
    if type(topo).__name__ != "Ramp":
        raise ValueError("engine: RAMP topology required")
    if topo.num_channels != 1:
        raise ValueError("engine: num_channels must be 1 (multi-channel dep "
                         "placement draws RNG per flow)")
    if topo.num_workers != topo.num_nodes:
        raise ValueError("engine: 1 worker per server required")
    if len(topo.worker_types) != 1:
        raise ValueError("engine: homogeneous cluster required")
    gen = cluster.jobs_generator
    mode = gen.job_sampler.sampling_mode
    infinite = mode in ("remove_and_repeat", "replace")
    if infinite and not np.isfinite(env.max_simulation_run_time):
        raise ValueError("engine: infinite pool needs finite "
                         "max_simulation_run_time")

    C, R, S = topo.shape
    W = topo.num_nodes
    A = env.max_partitions_per_op + 1
    spec = EngineSpec(
        C=C, R=R, S=S, W=W, A=A,
        eps=cluster.machine_epsilon,
        max_sim=float(env.max_simulation_run_time),
        mem_capacity=float(cluster.workers[0].memory_capacity),
        infinite_pool=infinite,
        static_shape_ok=_static_mask_ok(A, topo.shape),
        reward=_reward_spec_from_env(env),
        max_running=min(W, 256),
        device_type=cluster.device_type,
    )
    spec.shape_ptr, spec.shapes = _shape_lists(A - 1, topo.shape)

    device_type = cluster.device_type
    # one prototype job per model, pool order (stable model ids)
    model_to_job = {}
    for j in gen.job_sampler.original_pool:
        model_to_job.setdefault(j.details["model"], j)
    spec._name_to_id = {}

    # representative steps count (uniform across the pool by construction)
    steps_set = {j.num_training_steps for j in gen.job_sampler.original_pool}
    if len(steps_set) != 1:
        raise ValueError("engine: per-job num_training_steps must be uniform")

    saved_queue = dict(cluster.job_queue.jobs)
    saved_idx_maps = (dict(cluster.job_id_to_job_idx),
                      dict(cluster.job_idx_to_job_id))
    fake_idx = 20_000_000
    prepared, keys = [], []
    try:
        for model, proto in model_to_job.items():
            import copy as _copy
            mid = len(spec.models)
            spec._name_to_id[model] = mid
            g = proto.graph
            cc = g.compute_cost[device_type]

            # placement program inputs: topo order of forward ops + parents
            order = g.topo_order_fwd_subgraph()
            pos = {int(i): k for k, i in enumerate(order)}
            par_ptr, par_idx = [0], []
            op_mem = []
            for i in order:
                op_mem.append(float(g.memory_cost[i]))
                for e in g.in_edges_of(int(i)):
                    u = int(g.src[e])
                    if g.pass_type[u] == FWD:
                        par_idx.append(pos[u])
                par_ptr.append(len(par_idx))

            # action -> degree (max over per-op sip-ml splits); odd actions
            # > 1 are never mask-valid and get no entry
            a2d = np.zeros(A, dtype=np.int32)
            for a in range(1, A):
                if a > 1 and a % 2 != 0:
                    continue
                deg = 1
                for i in range(g.n):
                    if g.pass_type[i] != FWD:
                        continue
                    deg = max(deg, sip_ml_num_partitions(
                        float(cc[i]), env.min_op_run_time_quantum,
                        max_partitions_per_op=a))
                a2d[a] = deg

            ms = ModelSpec(
                name=model, n=g.n, m=g.m,
                seq_total=float(
                    proto.details["job_sequential_completion_time"][device_type]),
                mem_total=float(proto.details["job_total_op_memory_cost"]),
                dep_total=float(proto.details["job_total_dep_size"]),
                gf_static=_graph_features_static(proto, cluster),
                node_features=env.observation_function._node_features(proto, cluster),
                edge_features=env.observation_function._edge_features(proto, cluster),
                edges_src=g.src.astype(np.int64),
                edges_dst=g.dst.astype(np.int64),
                action_to_degree=a2d,
                seq_len=len(order),
                op_mem=np.asarray(op_mem, dtype=np.float64),
                par_ptr=np.asarray(par_ptr, dtype=np.int32),
                par_idx=np.asarray(par_idx, dtype=np.int32),
                num_training_steps=int(proto.num_training_steps),
            )
            spec.models.append(ms)

            # per distinct degree: run the real pipeline on the empty cluster
            job = _copy.deepcopy(proto)
            job.details["job_idx"] = fake_idx
            job.original_job.details["job_idx"] = fake_idx
            cluster.job_queue.jobs.clear()
            cluster.job_queue.jobs[job.job_id] = job
            cluster.job_id_to_job_idx[job.job_id] = fake_idx
            cluster.job_idx_to_job_id[fake_idx] = job.job_id
            fake_idx += 1
            seen = set()
            for a in range(1, A):
                degree = int(a2d[a])
                if degree in seen or degree < 1:
                    continue
                seen.add(degree)
                partition_action = {}
                for i in range(g.n):
                    if g.pass_type[i] != FWD:
                        continue
                    num = sip_ml_num_partitions(
                        float(cc[i]), env.min_op_run_time_quantum,
                        max_partitions_per_op=max(a, 1))
                    partition_action[g.names[i]] = num
                    partition_action[g.names[int(g.counterpart[i])]] = num
                op_partition = OpPartition({job.job_id: partition_action},
                                           cluster=cluster)
                assert op_partition.job_id_to_max_partition_degree[job.job_id] \
                    == degree
                # splits along the program sequence
                fwd_splits = op_partition.job_id_to_forward_op_id_to_mp_splits[
                    job.job_id]
                splits = np.asarray(
                    [int(fwd_splits.get(g.names[int(i)], 1)) for i in order],
                    dtype=np.int32)
                md = MdSpec(model_id=mid, degree=degree, placeable=False,
                            splits=splits)
                op_placement = env.op_placer.get(op_partition=op_partition,
                                                 cluster=cluster)
                if job.job_id in op_placement.action:
                    placement = op_placement.action[job.job_id]
                    op_schedule = env.op_scheduler.get(
                        op_partition=op_partition, op_placement=op_placement,
                        cluster=cluster)
                    dep_placement = env.dep_placer.get(
                        op_partition=op_partition, op_placement=op_placement,
                        cluster=cluster)
                    dep_schedule = env.dep_scheduler.get(
                        op_partition=op_partition, dep_placement=dep_placement,
                        cluster=cluster)
                    pjob = op_partition.partitioned_jobs[job.job_id]
                    ok = mount_job_for_lookahead(
                        cluster, pjob, placement, op_schedule, dep_placement,
                        dep_schedule)
                    if ok:
                        md.placeable = True
                        md.n_workers = len(set(placement.values()))
                        chans = set()
                        for dchs in dep_placement.action.get(job.job_id,
                                                             {}).values():
                            chans.update(c for c in dchs if c is not None)
                        md.n_channels = len(chans)
                        pg = pjob.graph
                        flow_mask = pjob.dep_is_flow
                        md.flow_size = float(pg.size[flow_mask].sum())
                        md.pj_n, md.pj_m = pg.n, pg.m
                        md.pj_mem = float(pjob.job_total_operation_memory_cost)
                        md.pj_dep = float(pjob.job_total_dependency_size)
                        md.pj_seq = float(pjob.details[
                            "job_sequential_completion_time"][device_type])
                        prepared.append(pjob)
                        keys.append((mid, degree))
                spec.md_index[(mid, degree)] = len(spec.mds)
                spec.mds.append(md)
    finally:
        cluster.job_queue.jobs.clear()
        cluster.job_queue.jobs.update(saved_queue)
        cluster.job_id_to_job_idx, cluster.job_idx_to_job_id = saved_idx_maps

    # lookahead memo values (batched HIP kernel when a GPU is given)
    results = run_lookahead_batch(prepared, device=lookahead_device)
    for (mid, degree), pjob, (t, comp, comm, active_sum) in zip(
            keys, prepared, results):
        steps = pjob.num_training_steps
        spec.memo[(mid, degree)] = (t * steps, comm * steps, comp * steps,
                                    active_sum)
        if verbose:
            print(f"engine memo model={spec.models[mid].name} degree={degree}"
                  f" jct={t * steps:.4f}")
    return spec


# ---------------------------------------------------------------------------
# Episode schedules
# ---------------------------------------------------------------------------

@dataclass
class EpisodeSchedule:
    """Per-(env, episode) arrival schedule + obs-normalisation scalars."""
    model_id: np.ndarray      # [n] i32
    frac: np.ndarray          # [n] f64
    max_acceptable: np.ndarray  # [n] f64 (frac * seq_total, job-order exact)
    nominal_next: np.ndarray  # [n+1] f64: time_next_job_to_arrive after draw k
    n: int = 0
    # episode obs-normalisation scalars (jobs_params of this episode)
    p_min_acc: float = 0.0
    p_max_acc: float = 0.0
    p_min_frac: float = 0.0
    p_max_frac: float = 0.0


def drain_episode_schedule(gen, spec: EngineSpec, seed: Optional[int],
                           reset_generator: bool = True) -> EpisodeSchedule:
    """Reproduce the env's per-episode RNG flow: seed_everything(seed) ->
    generator reset (frac resample) -> the exact _get_next_job draw sequence
    (``environment.py:110-180,207-228``), without building Job objects.

    Draw k (0-based; draw 0 happens at cluster.reset) fires when the sim
    clock satisfies t + eps >= nominal_next[k]; after it,
    time_next_job_to_arrive == nominal_next[k+1].  Draws with
    nominal_next[k] > max_sim + eps can never fire, so the drain stops there;
    over-draining is harmless because every episode re-seeds the RNG.
    """
    if seed is not None:
        seed_everything(seed)
    if reset_generator:
        gen.reset()
    p = gen.jobs_params
    models, fracs, accs = [], [], []
    nominal = [0.0]            # nominal_next[k] = threshold for draw k
    t_next = 0.0
    while len(gen) > 0 and t_next <= spec.max_sim + spec.eps:
        proto = gen.job_sampler.sample_ref()
        models.append(spec.model_id(proto.details["model"]))
        frac = proto.max_acceptable_job_completion_time_frac
        fracs.append(frac)
        seq = proto.details["job_sequential_completion_time"][spec.device_type]
        accs.append(frac * seq)
        t_next = t_next + gen.sample_interarrival_time()  # inf if pool empty
        nominal.append(t_next)
    return EpisodeSchedule(
        model_id=np.asarray(models, dtype=np.int32),
        frac=np.asarray(fracs, dtype=np.float64),
        max_acceptable=np.asarray(accs, dtype=np.float64),
        nominal_next=np.asarray(nominal, dtype=np.float64),
        n=len(models),
        p_min_acc=float(p["min_max_acceptable_job_completion_times"]),
        p_max_acc=float(p["max_max_acceptable_job_completion_times"]),
        p_min_frac=float(p["min_max_acceptable_job_completion_time_fracs"]),
        p_max_frac=float(p["max_max_acceptable_job_completion_time_fracs"]),
    )


# ---------------------------------------------------------------------------
# SoA state + CPU mirror step (the parity oracle for the HIP kernel)
# ---------------------------------------------------------------------------

class EngineState:
    """SoA state for B envs (numpy here; the GPU engine mirrors this layout
    in device tensors)."""

    def __init__(self, spec: EngineSpec, B: int, n_jobs_cap: int):
        K = spec.max_running
        self.B, self.K, self.n_jobs_cap = B, K, n_jobs_cap
        WW = (spec.W + 63) // 64
        self.WW = WW
        self.t = np.zeros(B)
        self.next_arrive = np.zeros(B)
        self.arr_ptr = np.zeros(B, dtype=np.int32)
        self.queued = np.full(B, -1, dtype=np.int32)
        self.n_running = np.zeros(B, dtype=np.int32)
        self.slot_md = np.full((B, K), -1, dtype=np.int32)
        self.slot_sched = np.full((B, K), -1, dtype=np.int32)
        self.slot_start = np.zeros((B, K))
        self.slot_jct = np.zeros((B, K))
        self.occ = np.zeros((B, WW), dtype=np.uint64)
        self.slot_occ = np.zeros((B, K, WW), dtype=np.uint64)
        self.snapshot = np.zeros(B, dtype=np.int32)
        self.ep_return = np.zeros(B)
        self.ep_len = np.zeros(B, dtype=np.int32)
        self.done = np.zeros(B, dtype=bool)
        self.status = np.full(B, ST_IDLE, dtype=np.int32)
        # per-job log
        self.log_status = np.full((B, n_jobs_cap), PENDING, dtype=np.uint8)
        self.log_md = np.full((B, n_jobs_cap), -1, dtype=np.int32)
        self.log_t_arr = np.zeros((B, n_jobs_cap))
        self.log_t_end = np.zeros((B, n_jobs_cap))
        self.log_order = np.full((B, n_jobs_cap), -1, dtype=np.int32)
        self.order_counter = np.zeros(B, dtype=np.int32)
        # obs outputs
        self.obs_model = np.full(B, -1, dtype=np.int32)
        self.obs_sched = np.full(B, -1, dtype=np.int32)
        self.obs_gf = np.zeros((B, 17), dtype=np.float32)
        self.obs_mask = np.zeros((B, spec.A), dtype=np.float32)
        # per-step outputs
        self.reward = np.zeros(B)
        self.step_done = np.zeros(B, dtype=bool)

    def reset_env(self, spec: EngineSpec, b: int, sched: EpisodeSchedule):
        """Mirror of cluster.reset + first _get_next_job (draw 0 at t=0)."""
        self.t[b] = 0.0
        self.arr_ptr[b] = 1
        self.next_arrive[b] = sched.nominal_next[1]
        self.queued[b] = 0
        self.n_running[b] = 0
        self.slot_md[b, :] = -1
        self.occ[b, :] = 0
        self.slot_occ[b, :, :] = 0
        self.snapshot[b] = 0
        self.ep_return[b] = 0.0
        self.ep_len[b] = 0
        self.done[b] = False
        self.status[b] = ST_IDLE
        n = sched.n
        self.log_status[b, :] = PENDING
        self.log_md[b, :] = -1
        self.log_t_arr[b, :n] = 0.0
        self.log_t_end[b, :n] = 0.0
        self.log_order[b, :] = -1
        self.order_counter[b] = 0
        self.log_t_arr[b, 0] = 0.0
        self._write_obs(spec, b, sched)

    # ------------------------------------------------------------------
    def _write_obs(self, spec: EngineSpec, b: int, sched: EpisodeSchedule):
        k = self.queued[b]
        if k < 0:
            return
        gf, mask, mid = compute_obs(spec, sched, int(k),
                                    int(self.snapshot[b]),
                                    int(self.n_running[b]))
        self.obs_gf[b] = gf
        self.obs_mask[b] = mask
        self.obs_model[b] = mid
        self.obs_sched[b] = k


def compute_obs(spec: EngineSpec, sched: EpisodeSchedule, k: int,
                snapshot: int, n_running: int):
    """Observation encode for queued job k (mirror of
    ``RampJobPartitioningObservation._graph_features`` dynamic entries +
    ``get_action_set_and_action_mask``)."""
    mid = int(sched.model_id[k])
    ms = spec.models[mid]
    gf = ms.gf_static.copy()
    acc = sched.max_acceptable[k]
    frac = sched.frac[k]
    lo, hi = sched.p_min_acc, sched.p_max_acc
    gf[3] = (acc - lo) / (hi - lo) if hi - lo != 0 else 1.0
    lo, hi = sched.p_min_frac, sched.p_max_frac
    gf[4] = (frac - lo) / (hi - lo) if hi - lo != 0 else 1.0
    gf[5] = frac
    gf[15] = snapshot / spec.W
    gf[16] = n_running / spec.W
    f32 = gf.astype(np.float32)
    f32[f32 < 0] += spec.eps   # numpy f32 += f64 scalar (obs fn order)
    num_avail = spec.W - snapshot
    mask = np.zeros(spec.A, dtype=np.float32)
    mask[0] = 1.0
    for a in range(1, spec.A):
        if spec.static_shape_ok[a] and a <= num_avail:
            mask[a] = 1.0
    return f32, mask, mid


def _block_servers(spec: EngineSpec, Cs: int, Rs: int, Ss: int,
                   origin) -> Optional[List[int]]:
    """get_block mirror (``agents/placement_utils.py:57-73``), returning dense
    server ids; None entry semantics (out-of-range diagonal wrap) -> None."""
    i, j, k = origin
    out = []
    if Ss == -1:
        for n in range(Cs):
            c = (i + n) % (spec.C + 1)
            r = (j + n) % (spec.R + 1)
            s = k % spec.S
            if c >= spec.C or r >= spec.R:
                return None     # dict-miss in the reference => invalid block
            out.append((c * spec.R + r) * spec.S + s)
    else:
        for c in range(Cs):
            for r in range(Rs):
                for s in range(Ss):
                    cc = (i + c) % spec.C
                    rr = (j + r) % spec.R
                    ss = (k + s) % spec.S
                    out.append((cc * spec.R + rr) * spec.S + ss)
    return out


def _occ_get(occ_row: np.ndarray, s: int) -> bool:
    return bool((occ_row[s >> 6] >> np.uint64(s & 63)) & np.uint64(1))


def _check_block(spec, occ_row, free_mem, block, op_size) -> bool:
    """check_block mirror (one-job-per-server + memory)."""
    if not block:
        return False
    for s in block:
        if _occ_get(occ_row, s):
            return False
        if free_mem[s] < op_size:
            return False
    return True


def _find_sub_block(spec: EngineSpec, occ_row, free_mem, split: int,
                    op_size: float) -> Optional[List[int]]:
    """find_sub_block/ff_block mirror: first fit over candidate shapes x
    origins (``placement_utils.py:132-158``)."""
    for si in range(spec.shape_ptr[split], spec.shape_ptr[split + 1]):
        Cs, Rs, Ss = (int(spec.shapes[si, 0]), int(spec.shapes[si, 1]),
                      int(spec.shapes[si, 2]))
        I = spec.C - Cs + 1
        J = spec.R - Rs + 1
        K = spec.S - Ss + 1
        if I <= 0 or J <= 0 or K <= 0:
            continue
        for i in range(I):
            for j in range(J):
                for k in range(K):
                    block = _block_servers(spec, Cs, Rs, Ss, (i, j, k))
                    if block is not None and _check_block(
                            spec, occ_row, free_mem, block, op_size):
                        return block
    return None


def _search_placement(spec: EngineSpec, ms: ModelSpec, md: MdSpec,
                      occ_row) -> Optional[np.ndarray]:
    """allocate() mirror (``placement_utils.py:268-283``): per-op parent
    placement then regular first-fit.  Returns the job's server-id union
    (sorted) or None."""
    free_mem = np.full(spec.W, spec.mem_capacity)
    op_servers: List[List[int]] = [[] for _ in range(ms.seq_len)]
    for k in range(ms.seq_len):
        split = int(md.splits[k])
        req = float(ms.op_mem[k])
        placed = False
        for pi in range(ms.par_ptr[k], ms.par_ptr[k + 1]):
            servers = op_servers[int(ms.par_idx[pi])]
            if split != len(servers):
                continue
            avail = 0.0
            for s in servers:
                avail += free_mem[s]
            if avail >= req:
                for s in servers:
                    free_mem[s] -= req / split
                    op_servers[k].append(s)
                placed = True
                break
        if placed:
            continue
        if split > spec.W:
            return None
        op_size = req / split
        block = _find_sub_block(spec, occ_row, free_mem, split, op_size)
        if block is None:
            return None
        for s in block:
            free_mem[s] -= op_size
            op_servers[k].append(s)
    union = sorted({s for servers in op_servers for s in servers})
    return np.asarray(union, dtype=np.int64)


def _jct_reward(spec: EngineSpec, value: float, seq_jct: float,
                blocked: bool, norm_seq: float = None) -> float:
    """LookaheadJobCompletionTime mirror (``envs/rewards.py``).  The
    normaliser divides by the seq JCT of the job the reward fn looked up:
    the PARTITIONED job's when placed, the original's when blocked."""
    r = spec.reward
    if blocked:
        base = (r.fail_const if r.fail_const is not None else seq_jct)
        reward = base * r.fail_factor
    else:
        reward = value
    if norm_seq is None:
        norm_seq = seq_jct
    if r.normaliser != 0 and reward != 0:
        den = norm_seq if r.normaliser == 1 else norm_seq * r.fail_factor
        reward = reward / den
    if r.inverse and reward != 0:
        reward = 1 / reward
    reward *= r.sign
    if r.transform_with_log:
        sign = math.copysign(1, reward)
        reward = sign * math.log(1 + abs(reward), 10)
    return reward


def cpu_step_env(spec: EngineSpec, st: EngineState, b: int,
                 sched: EpisodeSchedule, action: int,
                 memo: Optional[Dict] = None) -> int:
    """Step ONE env (mirror of RampJobPartitioningEnvironment.step).  Returns
    the env status (ST_OK, or ST_MISS leaving state untouched)."""
    if memo is None:
        memo = spec.memo
    if st.done[b] or st.queued[b] < 0:
        st.status[b] = ST_ERR
        return ST_ERR

    k = int(st.queued[b])
    mid = int(sched.model_id[k])
    ms = spec.models[mid]
    seq_jct = ms.seq_total
    placed = False
    blocked = False
    jct_total = 0.0

    if action != 0:
        degree = int(ms.action_to_degree[action])
        mdi = spec.md_index.get((mid, degree))
        md = spec.mds[mdi] if mdi is not None else None
        union = None
        if md is not None:
            ckey = (mdi, st.occ[b].tobytes())
            cache = spec.place_cache
            if ckey in cache:
                union = cache[ckey]
            else:
                union = _search_placement(spec, ms, md, st.occ[b])
                if len(cache) > 200000:
                    cache.clear()
                cache[ckey] = tuple(union) if union is not None else None
        if union is not None:
            entry = memo.get((mid, degree))
            if entry is None:
                st.status[b] = ST_MISS
                return ST_MISS
            jct_total = entry[0]
            if jct_total > sched.max_acceptable[k]:
                blocked = True          # lookahead contract violation
            else:
                placed = True
                # commit: occupy servers, append running slot
                slot = int(st.n_running[b])
                if slot >= st.K:
                    st.status[b] = ST_ERR
                    return ST_ERR
                st.slot_occ[b, slot, :] = 0
                for s in union:
                    w, bit = int(s) >> 6, np.uint64(1) << np.uint64(int(s) & 63)
                    st.occ[b, w] |= bit
                    st.slot_occ[b, slot, w] |= bit
                st.slot_md[b, slot] = mdi
                st.slot_sched[b, slot] = k
                st.slot_start[b, slot] = st.t[b]
                st.slot_jct[b, slot] = jct_total
                st.n_running[b] += 1
                st.log_status[b, k] = RUNNING
                st.log_md[b, k] = mdi
        else:
            blocked = True              # no feasible block placement
        if blocked:
            st.log_status[b, k] = BLOCKED
            st.log_md[b, k] = (mdi if (md is not None and union is not None)
                               else -1)
            st.log_t_end[b, k] = st.t[b]
            st.log_order[b, k] = st.order_counter[b]
            st.order_counter[b] += 1
    else:
        blocked = True                  # action 0: do not place
        st.log_status[b, k] = BLOCKED
        st.log_t_end[b, k] = st.t[b]
        st.log_order[b, k] = st.order_counter[b]
        st.order_counter[b] += 1
    st.queued[b] = -1

    # reward (the RL env extracts it straight after the FIRST cluster.step;
    # JCT/acceptance parts depend only on the placement decision, the
    # throughput family on that step's info_processed/step_time — which the
    # event loop below accrues tick by tick)
    r = spec.reward
    reward = 0.0
    if r.jct_weight != 0.0:
        norm_seq = (spec.mds[mdi].pj_seq if placed else seq_jct)
        reward += r.jct_weight * _jct_reward(spec, jct_total, seq_jct,
                                             not placed, norm_seq)
    if r.blocking_weight != 0.0:
        reward += r.blocking_weight * (r.acc_success if placed else r.acc_fail)
    st.ep_len[b] += 1
    tp_acc = 0.0
    tp_time = None          # set when the first cluster step closes
    t_step_start = float(st.t[b])

    # ---- outer event loop (+ idle fast-forward) ----
    eps = spec.eps
    while True:
        pool_empty = (not spec.infinite_pool) and math.isinf(st.next_arrive[b])
        tick = min(st.next_arrive[b] - st.t[b], spec.max_sim - st.t[b])
        for s in range(int(st.n_running[b])):
            elapsed = st.t[b] - st.slot_start[b, s]
            remaining = st.slot_jct[b, s] - elapsed
            tick = min(tick, remaining)
        st.snapshot[b] = int(sum(bin(int(w)).count("1") for w in st.occ[b]))
        if r.tp_kind and tp_time is None:
            # mirror of environment.py:658-674: per running job (mount
            # order), frac = tick/jct, info += quantity * frac — same f64
            # accumulation order, so the ratio is bitwise-equal
            for s_ in range(int(st.n_running[b])):
                frac = float(tick) / float(st.slot_jct[b, s_])
                md_ = spec.mds[int(st.slot_md[b, s_])]
                if r.tp_kind == 1:
                    tp_acc += md_.pj_mem * frac
                elif r.tp_kind == 2:
                    tp_acc += (md_.pj_mem + md_.pj_dep) * frac
                else:
                    ms_ = spec.models[md_.model_id]
                    tp_acc += (ms_.mem_total + ms_.dep_total) * frac
        st.t[b] = st.t[b] + tick
        # completions (in running order; removal preserves order)
        nr = int(st.n_running[b])
        keep = []
        for s in range(nr):
            elapsed = st.t[b] - st.slot_start[b, s]
            remaining = (st.slot_jct[b, s] - elapsed) - eps
            if remaining <= 0:
                kk = int(st.slot_sched[b, s])
                st.log_status[b, kk] = COMPLETED
                st.log_t_end[b, kk] = st.t[b]
                st.log_order[b, kk] = st.order_counter[b]
                st.order_counter[b] += 1
                st.occ[b] &= ~st.slot_occ[b, s]   # release servers
            else:
                keep.append(s)
        if len(keep) != nr:
            for i, s in enumerate(keep):
                st.slot_md[b, i] = st.slot_md[b, s]
                st.slot_sched[b, i] = st.slot_sched[b, s]
                st.slot_start[b, i] = st.slot_start[b, s]
                st.slot_jct[b, i] = st.slot_jct[b, s]
                st.slot_occ[b, i] = st.slot_occ[b, s]
            st.n_running[b] = len(keep)
        # arrival
        if not pool_empty:
            if (st.t[b] + eps) >= st.next_arrive[b]:
                kk = int(st.arr_ptr[b])
                st.log_t_arr[b, kk] = st.t[b]
                st.queued[b] = kk
                st.arr_ptr[b] += 1
                st.next_arrive[b] = sched.nominal_next[st.arr_ptr[b]]
        # done
        pool_empty = (not spec.infinite_pool) and math.isinf(st.next_arrive[b])
        done = (st.t[b] >= spec.max_sim) or (
            pool_empty and st.n_running[b] == 0 and st.queued[b] < 0)
        # a cluster step ends at the first COMPLETION or ARRIVAL (or done):
        # the RL env's reward covers only that first cluster step; later
        # iterations here belong to its idle fast-forward
        if (r.tp_kind and tp_time is None
                and (len(keep) != nr or st.queued[b] >= 0 or done)):
            tp_time = float(st.t[b]) - t_step_start
        if st.queued[b] >= 0 or done:
            break

    if r.tp_kind:
        step_time = (tp_time if tp_time is not None
                     else float(st.t[b]) - t_step_start)
        reward = (tp_acc / step_time
                  if (tp_acc != 0 and step_time != 0) else 0.0)
    st.reward[b] = reward
    st.ep_return[b] += reward

    if done:
        # finalise: still-running jobs are blocked (reference :1111-1121)
        for s in range(int(st.n_running[b])):
            kk = int(st.slot_sched[b, s])
            st.log_status[b, kk] = BLOCKED
            st.log_t_end[b, kk] = st.t[b]
            st.log_order[b, kk] = st.order_counter[b]
            st.order_counter[b] += 1
        st.done[b] = True
        st.step_done[b] = True
        st.status[b] = ST_OK
        return ST_OK

    st.step_done[b] = False
    st._write_obs(spec, b, sched)
    st.status[b] = ST_OK
    return ST_OK


# ---------------------------------------------------------------------------
# Episode stats reconstruction (host, from the per-job log)
# ---------------------------------------------------------------------------

def build_episode_stats(spec: EngineSpec, sched: EpisodeSchedule,
                        st: EngineState, b: int) -> Dict:
    """Reconstruct the cluster ``episode_stats`` vocabulary from the compact
    per-job log (reference ``ramp_cluster_environment.py:1046-1167``).

    Exact for every per-job list/count metric and for the info-processed
    totals (a job's per-tick fractional contributions telescope to
    (t_end - t_start)/jct x quantity).  The four mean-of-step-means iteration
    metrics (mean_num_jobs_running etc.) need per-event-loop-iteration data
    and are NOT reproduced; engine consumers get the exact core vocabulary.
    """
    n_arrived = int(st.arr_ptr[b])
    t_end = float(st.t[b])
    es: Dict = {
        "num_jobs_arrived": n_arrived,
        "episode_start_time": 0.0,
        "episode_end_time": t_end,
        "episode_time": t_end,
        "episode_return": float(st.ep_return[b]),
        "episode_len": int(st.ep_len[b]),
    }
    status = st.log_status[b, :n_arrived]
    order = st.log_order[b, :n_arrived]
    mds = st.log_md[b, :n_arrived]
    t_arr = st.log_t_arr[b, :n_arrived]
    t_fin = st.log_t_end[b, :n_arrived]
    models = sched.model_id[:n_arrived]

    comp_keys = ("job_completion_time", "job_completion_time_speedup",
                 "job_communication_overhead_time",
                 "job_computation_overhead_time", "jobs_completed_num_nodes",
                 "jobs_completed_num_edges",
                 "jobs_completed_total_operation_memory_cost",
                 "jobs_completed_total_dependency_size",
                 "jobs_completed_max_partitions_per_op",
                 "jobs_completed_job_sequential_completion_time",
                 "jobs_completed_max_acceptable_job_completion_time_frac",
                 "jobs_completed_max_acceptable_job_completion_time",
                 "jobs_completed_num_mounted_workers",
                 "jobs_completed_num_mounted_channels",
                 "jobs_completed_mean_mounted_worker_utilisation_frac",
                 "jobs_completed_original_demand_num_nodes",
                 "jobs_completed_original_demand_num_edges",
                 "jobs_completed_original_demand_total_operation_memory_cost",
                 "jobs_completed_original_demand_total_dependency_size")
    blk_keys = ("jobs_blocked_num_nodes", "jobs_blocked_num_edges",
                "jobs_blocked_total_operation_memory_cost",
                "jobs_blocked_total_dependency_size",
                "jobs_blocked_job_sequential_completion_time",
                "jobs_blocked_max_acceptable_job_completion_time_frac",
                "jobs_blocked_max_acceptable_job_completion_time",
                "jobs_blocked_original_demand_num_nodes",
                "jobs_blocked_original_demand_num_edges",
                "jobs_blocked_original_demand_total_operation_memory_cost",
                "jobs_blocked_original_demand_total_dependency_size")
    for kk in comp_keys + blk_keys:
        es[kk] = []

    # registration order within each category matches the CPU env's
    completed = [k for k in range(n_arrived) if status[k] == COMPLETED]
    completed.sort(key=lambda k: order[k])
    blocked = [k for k in range(n_arrived) if status[k] == BLOCKED]
    blocked.sort(key=lambda k: order[k])

    info = {k: 0.0 for k in ("compute_info_processed", "dep_info_processed",
                             "flow_info_processed", "cluster_info_processed",
                             "demand_compute_info_processed",
                             "demand_dep_info_processed",
                             "demand_total_info_processed")}

    def _accumulate_info(md: MdSpec, frac_done: float, ms: ModelSpec):
        info["compute_info_processed"] += md.pj_mem * frac_done
        info["dep_info_processed"] += md.pj_dep * frac_done
        info["flow_info_processed"] += md.flow_size * frac_done
        info["cluster_info_processed"] += (md.pj_mem + md.pj_dep) * frac_done
        info["demand_compute_info_processed"] += ms.mem_total * frac_done
        info["demand_dep_info_processed"] += ms.dep_total * frac_done
        info["demand_total_info_processed"] += (
            (ms.mem_total + ms.dep_total) * frac_done)

    for k in completed:
        ms = spec.models[int(models[k])]
        md = spec.mds[int(mds[k])]
        entry = spec.memo[(md.model_id, md.degree)]
        jct_total, comm_oh, comp_oh, active_sum = entry
        jct = t_fin[k] - t_arr[k]
        es["job_completion_time"].append(jct)
        es["job_completion_time_speedup"].append(md.pj_seq / jct)
        es["job_communication_overhead_time"].append(comm_oh)
        es["job_computation_overhead_time"].append(comp_oh)
        es["jobs_completed_num_nodes"].append(md.pj_n)
        es["jobs_completed_num_edges"].append(md.pj_m)
        es["jobs_completed_total_operation_memory_cost"].append(md.pj_mem)
        es["jobs_completed_total_dependency_size"].append(md.pj_dep)
        es["jobs_completed_max_partitions_per_op"].append(md.degree)
        es["jobs_completed_job_sequential_completion_time"].append(md.pj_seq)
        es["jobs_completed_max_acceptable_job_completion_time_frac"].append(
            float(sched.frac[k]))
        # _register_completed_lookahead's reset_job RECOMPUTES max_acceptable
        # from the PARTITIONED immutable (frac x pj_seq) and its overlay does
        # not carry the original's value — the blocking DECISION, by
        # contrast, used the original's (OpPartition overlay quirk)
        es["jobs_completed_max_acceptable_job_completion_time"].append(
            float(sched.frac[k]) * md.pj_seq)
        es["jobs_completed_num_mounted_workers"].append(md.n_workers)
        es["jobs_completed_num_mounted_channels"].append(md.n_channels)
        es["jobs_completed_mean_mounted_worker_utilisation_frac"].append(
            active_sum / (md.n_workers * jct_total))
        es["jobs_completed_original_demand_num_nodes"].append(ms.n)
        es["jobs_completed_original_demand_num_edges"].append(ms.m)
        es["jobs_completed_original_demand_total_operation_memory_cost"].append(
            ms.mem_total)
        es["jobs_completed_original_demand_total_dependency_size"].append(
            ms.dep_total)
        _accumulate_info(md, (t_fin[k] - t_arr[k]) / jct_total, ms)

    for k in blocked:
        ms = spec.models[int(models[k])]
        es["jobs_blocked_num_nodes"].append(ms.n)
        es["jobs_blocked_num_edges"].append(ms.m)
        es["jobs_blocked_total_operation_memory_cost"].append(ms.mem_total)
        es["jobs_blocked_total_dependency_size"].append(ms.dep_total)
        es["jobs_blocked_job_sequential_completion_time"].append(ms.seq_total)
        es["jobs_blocked_max_acceptable_job_completion_time_frac"].append(
            float(sched.frac[k]))
        es["jobs_blocked_max_acceptable_job_completion_time"].append(
            float(sched.max_acceptable[k]))
        es["jobs_blocked_original_demand_num_nodes"].append(ms.n)
        es["jobs_blocked_original_demand_num_edges"].append(ms.m)
        es["jobs_blocked_original_demand_total_operation_memory_cost"].append(
            ms.mem_total)
        es["jobs_blocked_original_demand_total_dependency_size"].append(
            ms.dep_total)
        # end-of-episode-blocked jobs that RAN contribute partial progress
        if mds[k] >= 0 and t_fin[k] > t_arr[k]:
            md = spec.mds[int(mds[k])]
            jct_total = spec.memo[(md.model_id, md.degree)][0]
            _accumulate_info(md, (t_fin[k] - t_arr[k]) / jct_total, ms)

    es["num_jobs_completed"] = len(completed)
    es["num_jobs_blocked"] = len(blocked)
    es["blocking_rate"] = (len(blocked) / n_arrived) if n_arrived else 0
    es["acceptance_rate"] = (len(completed) / n_arrived) if n_arrived else 0
    es.update({k: float(v) for k, v in info.items()})
    for tp, key in (("mean_compute_throughput", "compute_info_processed"),
                    ("mean_dep_throughput", "dep_info_processed"),
                    ("mean_flow_throughput", "flow_info_processed"),
                    ("mean_cluster_throughput", "cluster_info_processed"),
                    ("mean_demand_compute_throughput",
                     "demand_compute_info_processed"),
                    ("mean_demand_dep_throughput", "demand_dep_info_processed"),
                    ("mean_demand_total_throughput",
                     "demand_total_info_processed")):
        es[tp] = (es[key] / t_end) if (es[key] != 0 and t_end != 0) else 0
    # load rates: (orig mem+dep) / (nominal_next_after_draw - draw_time)
    rates = []
    for k in range(n_arrived):
        ms = spec.models[int(models[k])]
        denom = sched.nominal_next[k + 1] - t_arr[k]
        rates.append((ms.mem_total + ms.dep_total) / denom)
    es["mean_load_rate"] = float(np.mean(rates)) if rates else 0
    return es


# ---------------------------------------------------------------------------
# CPU engine backend: the mirror as a production vectorised engine
# ---------------------------------------------------------------------------

class CpuEngine:
    """Drop-in CPU counterpart of ``gpu_engine.GpuEngine`` built on the
    (bitwise parity-tested) mirror ``cpu_step_env``: the same ``T`` tensor
    surface (zero-copy torch views of the numpy SoA state), ``step``,
    ``reset_env`` and ``episode_stats`` — so ``EngineVectorEnv`` runs at
    engine speed on CPU-only machines instead of requiring subprocess env
    workers.  No RampJobPartitioningEnvironment objects are stepped after
    spec compile; the per-env step is the compact-state mirror."""

    def __init__(self, spec: EngineSpec, B: int, device=None,
                 n_jobs_cap: int = 64, sch_cap: Optional[int] = None):
        import torch
        self.spec = spec
        self.B = B
        self.st = EngineState(spec, B=B, n_jobs_cap=n_jobs_cap)
        self.schedules: List[Optional[EpisodeSchedule]] = [None] * B
        st = self.st
        self.T = {
            "obs_gf": torch.from_numpy(st.obs_gf),
            "obs_mask": torch.from_numpy(st.obs_mask),
            "obs_model": torch.from_numpy(st.obs_model),
            "reward": torch.from_numpy(st.reward),
            "done": torch.from_numpy(st.done),
            "status": torch.from_numpy(st.status),
            "log_status": torch.from_numpy(st.log_status),
            "log_t_arr": torch.from_numpy(st.log_t_arr),
            "log_t_end": torch.from_numpy(st.log_t_end),
        }

    def reset_env(self, b: int, sched: EpisodeSchedule):
        if sched.n + 1 > self.st.n_jobs_cap:
            raise ValueError(
                f"CpuEngine: schedule length {sched.n} exceeds capacity "
                f"{self.st.n_jobs_cap} — construct with a larger n_jobs_cap")
        self.schedules[b] = sched
        self.st.reset_env(self.spec, b, sched)

    def step(self, actions, active: Optional[List[int]] = None):
        acts = actions.detach().cpu().numpy()
        idx = range(self.B) if active is None else active
        for b in idx:
            if self.st.done[b]:
                self.st.status[b] = 0
                continue
            status = cpu_step_env(self.spec, self.st, b,
                                  self.schedules[b], int(acts[b]))
            if status != ST_OK:
                raise RuntimeError(f"CpuEngine env {b}: status {status}")
        return self.T["status"]

    def episode_stats(self, b: int) -> Dict:
        return build_episode_stats(self.spec, self.schedules[b], self.st, b)
