"""GPU-resident vectorised env engine: device tensors + the env_step HIP
kernel (``ops/hip/env_step.hip``), host control plane for schedules, episode
resets and memo-miss servicing.

Layout contract: the tensor-list order here MUST match the ``enum`` in
``env_step.hip`` (T_*, I_*, F_*).  Semantics contract: the kernel is a
translation of ``vec_engine.cpu_step_env`` — the GPU parity test asserts
bitwise f64 state equality against the CPU mirror over whole episodes.
"""
from __future__ import annotations

from typing import Dict, List, Optional

import numpy as np
import torch

from .vec_engine import (EngineSpec, EpisodeSchedule, PENDING, ST_MISS, ST_OK,
                         ST_STEP, build_episode_stats, compute_obs)


def _pack_spec_tensors(spec: EngineSpec, device) -> Dict[str, torch.Tensor]:
    M = len(spec.models)
    D = len(spec.mds)
    A = spec.A
    SEQ_CAP = max(ms.seq_len for ms in spec.models)
    PAR_CAP = max(len(ms.par_idx) for ms in spec.models)
    model_gf = np.stack([ms.gf_static for ms in spec.models])
    model_seq = np.array([ms.seq_total for ms in spec.models])
    model_seq_len = np.array([ms.seq_len for ms in spec.models], dtype=np.int32)
    op_mem = np.zeros((M, SEQ_CAP))
    par_ptr = np.zeros((M, SEQ_CAP + 1), dtype=np.int32)
    par_idx = np.zeros((M, max(PAR_CAP, 1)), dtype=np.int32)
    model_a2md = np.full((M, A), -1, dtype=np.int32)
    for m, ms in enumerate(spec.models):
        op_mem[m, :ms.seq_len] = ms.op_mem
        par_ptr[m, :ms.seq_len + 1] = ms.par_ptr
        par_idx[m, :len(ms.par_idx)] = ms.par_idx
        for a in range(1, A):
            deg = int(ms.action_to_degree[a])
            if deg >= 1:
                mdi = spec.md_index.get((m, deg))
                if mdi is not None:
                    model_a2md[m, a] = mdi
    md_splits = np.zeros((max(D, 1), SEQ_CAP), dtype=np.int32)
    md_pjseq = np.zeros(max(D, 1))
    md_degree = np.zeros(max(D, 1), dtype=np.int32)
    md_model = np.zeros(max(D, 1), dtype=np.int32)
    for d, md in enumerate(spec.mds):
        seq_len = spec.models[md.model_id].seq_len
        md_splits[d, :seq_len] = md.splits
        md_pjseq[d] = md.pj_seq
        md_degree[d] = md.degree
        md_model[d] = md.model_id

    t = lambda a, dt: torch.as_tensor(np.ascontiguousarray(a), device=device,
                                      dtype=dt)
    return {
        "static_ok": t(spec.static_shape_ok, torch.uint8),
        "shape_ptr": t(spec.shape_ptr, torch.int32),
        "shapes": t(spec.shapes, torch.int32),
        "model_gf": t(model_gf, torch.float64),
        "model_seq": t(model_seq, torch.float64),
        "model_a2md": t(model_a2md, torch.int32),
        "model_seq_len": t(model_seq_len, torch.int32),
        "op_mem": t(op_mem, torch.float64),
        "par_ptr": t(par_ptr, torch.int32),
        "par_idx": t(par_idx, torch.int32),
        "md_splits": t(md_splits, torch.int32),
        "md_pjseq": t(md_pjseq, torch.float64),
        "md_degree": t(md_degree, torch.int32),
        "md_model": t(md_model, torch.int32),
        "_SEQ_CAP": SEQ_CAP, "_PAR_CAP": max(PAR_CAP, 1),
    }


class _LogView:
    """Host view of one env's state/log rows, shaped like EngineState rows so
    ``build_episode_stats(spec, sched, view, 0)`` works unchanged."""

    def __init__(self, eng: "GpuEngine", b: int):
        n = eng.NJOBS
        self.arr_ptr = np.array([int(eng.T["arr_ptr"][b])], dtype=np.int32)
        self.t = np.array([float(eng.T["t"][b])])
        self.ep_return = np.array([float(eng.T["ep_return"][b])])
        self.ep_len = np.array([int(eng.T["ep_len"][b])], dtype=np.int32)
        self.log_status = eng.T["log_status"][b].cpu().numpy()[None]
        self.log_md = eng.T["log_md"][b].cpu().numpy()[None]
        self.log_t_arr = eng.T["log_t_arr"][b].cpu().numpy()[None]
        self.log_t_end = eng.T["log_t_end"][b].cpu().numpy()[None]
        self.log_order = eng.T["log_order"][b].cpu().numpy()[None]


class GpuEngine:
    """B vectorised envs resident on one GPU; one kernel launch per step."""

    def __init__(self, spec: EngineSpec, B: int, device,
                 n_jobs_cap: Optional[int] = None, hash_size: int = 1024,
                 preload_memo: bool = True):
        from .. import ops as hip_ops
        self.ext = hip_ops.get_extension(required=True)
        self.spec = spec
        self.B = B
        self.device = device
        self.K = spec.max_running
        self.WW = (spec.W + 63) // 64
        assert self.WW <= 16, "engine supports up to 1024 workers"
        assert hash_size & (hash_size - 1) == 0
        self.HS = hash_size
        self.NJOBS = n_jobs_cap or 0       # grown at first schedule
        self.SCH = 0
        self.T: Dict[str, torch.Tensor] = _pack_spec_tensors(spec, device)
        self.schedules: List[Optional[EpisodeSchedule]] = [None] * B
        self._hash_slots: Dict[int, int] = {}
        self._alloc_hash()
        if preload_memo:
            for (mid, deg), vals in spec.memo.items():
                self.hash_insert(mid, deg, vals)
        self._state_alloc_done = False

    # ------------------------------------------------------------------
    def _alloc_hash(self):
        dev = self.device
        self.T["hash_keys"] = torch.full((self.HS,), -1, dtype=torch.int64,
                                         device=dev)
        self.T["hash_vals"] = torch.zeros((self.HS, 4), dtype=torch.float64,
                                          device=dev)

    def hash_insert(self, model_id: int, degree: int, vals):
        """Open-addressing insert (host-side mirror keeps slot bookkeeping)."""
        key = (int(model_id) << 20) | int(degree)
        slot = (key * 0x9E3779B97F4A7C15 >> 40) & (self.HS - 1)
        while True:
            owner = self._hash_slots.get(slot)
            if owner is None or owner == key:
                break
            slot = (slot + 1) & (self.HS - 1)
        self._hash_slots[slot] = key
        self.T["hash_keys"][slot] = key
        self.T["hash_vals"][slot] = torch.as_tensor(
            np.asarray(vals, dtype=np.float64), device=self.device)

    # ------------------------------------------------------------------
    def _alloc_state(self, sch_cap: int, njobs_cap: int):
        B, K, WW, A = self.B, self.K, self.WW, self.spec.A
        dev = self.device
        z = lambda shape, dt: torch.zeros(shape, dtype=dt, device=dev)
        self.SCH = sch_cap
        self.NJOBS = njobs_cap
        self.T.update({
            "sch_model": z((B, sch_cap), torch.int32),
            "sch_frac": z((B, sch_cap), torch.float64),
            "sch_acc": z((B, sch_cap), torch.float64),
            "sch_nominal": z((B, sch_cap + 1), torch.float64),
            "sch_n": z((B,), torch.int32),
            "sch_params": z((B, 4), torch.float64),
            "t": z((B,), torch.float64),
            "next_arrive": z((B,), torch.float64),
            "arr_ptr": z((B,), torch.int32),
            "queued": torch.full((B,), -1, dtype=torch.int32, device=dev),
            "n_running": z((B,), torch.int32),
            "slot_md": torch.full((B, K), -1, dtype=torch.int32, device=dev),
            "slot_sched": torch.full((B, K), -1, dtype=torch.int32, device=dev),
            "slot_start": z((B, K), torch.float64),
            "slot_jct": z((B, K), torch.float64),
            "slot_occ": z((B, K, WW), torch.int64),
            "occ": z((B, WW), torch.int64),
            "snapshot": z((B,), torch.int32),
            "ep_return": z((B,), torch.float64),
            "ep_len": z((B,), torch.int32),
            "done": z((B,), torch.uint8),
            "status": z((B,), torch.int32),
            "log_status": z((B, njobs_cap), torch.uint8),
            "log_md": torch.full((B, njobs_cap), -1, dtype=torch.int32,
                                 device=dev),
            "log_t_arr": z((B, njobs_cap), torch.float64),
            "log_t_end": z((B, njobs_cap), torch.float64),
            "log_order": torch.full((B, njobs_cap), -1, dtype=torch.int32,
                                    device=dev),
            "order_counter": z((B,), torch.int32),
            "obs_model": torch.full((B,), -1, dtype=torch.int32, device=dev),
            "obs_sched": torch.full((B,), -1, dtype=torch.int32, device=dev),
            "obs_gf": z((B, 17), torch.float32),
            "obs_mask": z((B, A), torch.float32),
            "reward": z((B,), torch.float64),
            "step_done": z((B,), torch.uint8),
            "actions": z((B,), torch.int32),
        })
        self._state_alloc_done = True

    # ------------------------------------------------------------------
    def reset_env(self, b: int, sched: EpisodeSchedule):
        """Upload the episode schedule + mirror of EngineState.reset_env."""
        if not self._state_alloc_done:
            cap = max(self.NJOBS, sched.n + 8)
            self._alloc_state(sch_cap=cap, njobs_cap=cap)
        if sched.n + 1 > self.SCH:
            raise RuntimeError(
                f"schedule length {sched.n} exceeds engine capacity "
                f"{self.SCH}; size n_jobs_cap accordingly")
        dev = self.device
        self.schedules[b] = sched
        T = self.T
        n = sched.n
        T["sch_model"][b, :n] = torch.as_tensor(sched.model_id, device=dev)
        T["sch_frac"][b, :n] = torch.as_tensor(sched.frac, device=dev)
        T["sch_acc"][b, :n] = torch.as_tensor(sched.max_acceptable, device=dev)
        T["sch_nominal"][b, :n + 1] = torch.as_tensor(sched.nominal_next,
                                                      device=dev)
        T["sch_n"][b] = n
        T["sch_params"][b] = torch.as_tensor(
            [sched.p_min_acc, sched.p_max_acc, sched.p_min_frac,
             sched.p_max_frac], dtype=torch.float64, device=dev)
        T["t"][b] = 0.0
        T["arr_ptr"][b] = 1
        T["next_arrive"][b] = float(sched.nominal_next[1])
        T["queued"][b] = 0
        T["n_running"][b] = 0
        T["slot_md"][b] = -1
        T["occ"][b] = 0
        T["slot_occ"][b] = 0
        T["snapshot"][b] = 0
        T["ep_return"][b] = 0.0
        T["ep_len"][b] = 0
        T["done"][b] = 0
        T["status"][b] = 0
        T["log_status"][b] = PENDING
        T["log_md"][b] = -1
        T["log_t_arr"][b] = 0.0
        T["log_t_end"][b] = 0.0
        T["log_order"][b] = -1
        T["order_counter"][b] = 0
        gf, mask, mid = compute_obs(self.spec, sched, 0, 0, 0)
        T["obs_gf"][b] = torch.as_tensor(gf, device=dev)
        T["obs_mask"][b] = torch.as_tensor(mask, device=dev)
        T["obs_model"][b] = mid
        T["obs_sched"][b] = 0

    # ------------------------------------------------------------------
    _ORDER = ("static_ok", "shape_ptr", "shapes", "model_gf", "model_seq",
              "model_a2md", "model_seq_len", "op_mem", "par_ptr", "par_idx",
              "md_splits", "md_pjseq", "md_degree", "md_model",
              "hash_keys", "hash_vals",
              "sch_model", "sch_frac", "sch_acc", "sch_nominal", "sch_n",
              "sch_params",
              "t", "next_arrive", "arr_ptr", "queued", "n_running",
              "slot_md", "slot_sched", "slot_start", "slot_jct", "slot_occ",
              "occ", "snapshot", "ep_return", "ep_len", "done", "status",
              "log_status", "log_md", "log_t_arr", "log_t_end", "log_order",
              "order_counter", "obs_model", "obs_sched", "obs_gf", "obs_mask",
              "reward", "step_done", "actions")

    def _scalars(self):
        spec = self.spec
        r = spec.reward
        if getattr(r, "tp_kind", 0):
            raise ValueError(
                "GpuEngine: throughput rewards are CPU-mirror-only for now "
                "(env_step.hip port pending) — use the CPU env path")
        iscal = [self.B, spec.C, spec.R, spec.S, spec.W, self.WW, spec.A,
                 self.K, self.T["_SEQ_CAP"], self.T["_PAR_CAP"], self.SCH,
                 self.NJOBS, self.HS, int(spec.infinite_pool),
                 int(r.inverse), int(r.transform_with_log), int(r.normaliser),
                 int(r.fail_const is None)]
        fscal = [spec.eps, spec.max_sim, spec.mem_capacity, r.sign,
                 r.fail_factor,
                 0.0 if r.fail_const is None else r.fail_const,
                 r.acc_success, r.acc_fail, r.jct_weight, r.blocking_weight]
        return fscal, iscal

    def step(self, actions: torch.Tensor, active: Optional[List[int]] = None,
             max_miss_rounds: int = 8):
        """One batched env step.  ``actions`` int32 [B] on device.  Launches
        the kernel; services memo misses (insert + relaunch).  Statuses left
        in T['status'] (ST_OK / ST_ERR)."""
        T = self.T
        T["actions"].copy_(actions.to(torch.int32))
        if active is None:
            T["status"].fill_(ST_STEP)
            T["status"][T["done"] != 0] = 0
        else:
            T["status"].fill_(0)
            for b in active:
                T["status"][b] = ST_STEP
        fscal, iscal = self._scalars()
        for _ in range(max_miss_rounds):
            self.ext.env_step_batch([T[k] for k in self._ORDER], fscal, iscal)
            status = T["status"].cpu().numpy()
            miss = np.flatnonzero(status == ST_MISS)
            if len(miss) == 0:
                break
            acts = T["actions"].cpu().numpy()
            for b in miss:
                mid = int(T["obs_model"][b])
                deg = int(self.spec.models[mid].action_to_degree[acts[b]])
                vals = self.spec.memo.get((mid, deg))
                if vals is None:
                    raise RuntimeError(
                        f"memo miss for (model {mid}, degree {deg}) with no "
                        "compiled lookahead — spec.memo incomplete")
                self.hash_insert(mid, deg, vals)
                T["status"][int(b)] = ST_STEP
        return T["status"]

    # ------------------------------------------------------------------
    def episode_stats(self, b: int) -> Dict:
        view = _LogView(self, b)
        return build_episode_stats(self.spec, self.schedules[b], view, 0)
