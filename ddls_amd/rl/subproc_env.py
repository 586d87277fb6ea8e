"""Multiprocess vectorised envs.

Replaces the reference's Ray rollout workers (RLlib ``num_workers: 8``,
SURVEY.md K6/K8) with fork-based worker processes over pipes: each worker owns
a shard of envs, steps them on request and ships back compact observations
(a few KB each) — no object store, no RPC framework.
"""
from __future__ import annotations

import multiprocessing as mp
import os
from typing import List, Optional

import numpy as np

from .rollout import CompactObs


def _worker_loop(conn, env_fn_count: int, env_fn, base_seed: int,
                 shm_buf=None):
    envs = [env_fn() for _ in range(env_fn_count)]
    episode_counters = [0] * len(envs)
    episode_returns = np.zeros(len(envs))
    episode_lens = np.zeros(len(envs), dtype=np.int64)
    policy = None          # lazy CPU policy copy for worker-side rollouts
    obs_cache = None
    rng = None
    shm_views = None       # zero-copy param views into the fork-shared buffer
    compact_memo = {}      # env obs dict (cached per (model,frac)) -> CompactObs

    def to_compact(o):
        key = id(o)
        hit = compact_memo.get(key)
        if hit is not None and hit[0] is o:
            return hit[1]
        if len(compact_memo) > 8192:
            compact_memo.clear()
        co = CompactObs.from_obs(o)
        compact_memo[key] = (o, co)
        return co
    try:
        while True:
            cmd, payload = conn.recv()
            if cmd == "shm_meta":
                # parent laid out flat params in the fork-inherited RawArray;
                # build numpy views once — every later rollout sees the
                # freshest weights with NO weight pickling at all
                raw = np.frombuffer(shm_buf, dtype=np.uint8)
                shm_views = {}
                for key, off, nbytes, shape, dtype in payload:
                    shm_views[key] = (raw[off:off + nbytes]
                                      .view(np.dtype(dtype)).reshape(shape))
                policy = None
                conn.send("ok")
                continue
            if cmd == "rollout":
                # RLlib-style worker-side rollout: the worker holds a numpy
                # inference copy of the policy (torch op-dispatch dominates at
                # these tiny shapes) and runs T steps locally — one IPC round
                # trip per iteration instead of per env step.
                from ..models.numpy_policy import NumpyGNNPolicy
                state_dict, steps, policy_cfg = payload
                if rng is None:
                    rng = np.random.RandomState((base_seed * 7919 + 13)
                                                % (2 ** 31))
                num_actions = len(envs[0].action_set)
                if state_dict is None:
                    # shared-memory path: the policy's arrays ARE the shm
                    # views, so reuse it as-is
                    if policy is None:
                        policy = NumpyGNNPolicy(shm_views, policy_cfg,
                                                num_actions)
                        # the zero-copy contract must hold or rollout weights
                        # would freeze at first-build values
                        assert policy.aliases_inputs, (
                            "shared-memory policy views were copied, not "
                            "aliased — non-fp32/non-contiguous param?")
                else:
                    policy = NumpyGNNPolicy(state_dict, policy_cfg,
                                            num_actions)
                if obs_cache is None:
                    obs_cache = [to_compact(env.reset(seed=base_seed + 1000 * i))
                                 for i, env in enumerate(envs)]
                traj = {"obs": [], "actions": [], "logp": [], "values": [],
                        "rewards": [], "dones": [], "lp_all": []}
                stats_out = []
                n = len(envs)
                num_actions = len(envs[0].action_set)
                for _t in range(steps):
                    actions = np.zeros(n, dtype=np.int64)
                    logps = np.zeros(n, dtype=np.float32)
                    values = np.zeros(n, dtype=np.float32)
                    lp_all = np.zeros((n, num_actions), dtype=np.float32)
                    for i, o in enumerate(obs_cache):
                        actions[i], logps[i], values[i], lp_all[i] = \
                            policy.act_full(o, rng)
                    traj["lp_all"].append(lp_all)
                    traj["obs"].append(list(obs_cache))
                    traj["actions"].append(actions)
                    traj["logp"].append(logps)
                    traj["values"].append(values)
                    rewards = np.zeros(n)
                    dones = np.zeros(n, dtype=bool)
                    for i, env in enumerate(envs):
                        o, r, done, _ = env.step(int(actions[i]))
                        rewards[i] = r
                        dones[i] = done
                        episode_returns[i] += r
                        episode_lens[i] += 1
                        if done:
                            st = dict(env.cluster.episode_stats)
                            st["episode_return"] = float(episode_returns[i])
                            st["episode_len"] = int(episode_lens[i])
                            stats_out.append(st)
                            episode_returns[i] = 0.0
                            episode_lens[i] = 0
                            episode_counters[i] += 1
                            o = env.reset(seed=base_seed + 1000 * i
                                          + episode_counters[i])
                        obs_cache[i] = to_compact(o)
                    traj["rewards"].append(rewards)
                    traj["dones"].append(dones)
                # bootstrap values of the final obs
                traj["bootstrap_values"] = np.array(
                    [policy.forward(o)[1] for o in obs_cache],
                    dtype=np.float32)
                conn.send((traj, stats_out))
                continue
            if cmd == "warm":
                # deterministically exercise each env's (model, action) space
                # so the empty-cluster pipeline caches (which survive resets)
                # are hot before the first timed rollout; envs are re-reset
                # afterwards so trajectories are unchanged
                steps = payload
                for i, env in enumerate(envs):
                    o = env.reset(seed=base_seed + 1000 * i)
                    for t in range(steps):
                        mask = o["action_mask"].astype(bool)
                        valid = o["action_set"][mask]
                        o, _r, done, _ = env.step(int(valid[t % len(valid)]))
                        if done:
                            o = env.reset(seed=base_seed + 1000 * i)
                obs_cache = None
                episode_returns[:] = 0.0
                episode_lens[:] = 0
                conn.send("ok")
                continue
            if cmd == "reset":
                obs = []
                for i, env in enumerate(envs):
                    o = env.reset(seed=base_seed + 1000 * i)
                    obs.append(to_compact(o))
                obs_cache = list(obs)
                conn.send(obs)
            elif cmd == "step":
                actions = payload
                obs_out, rewards, dones, stats_out = [], [], [], []
                for i, env in enumerate(envs):
                    o, r, done, _ = env.step(int(actions[i]))
                    episode_returns[i] += r
                    episode_lens[i] += 1
                    if done:
                        stats = dict(env.cluster.episode_stats)
                        stats["episode_return"] = float(episode_returns[i])
                        stats["episode_len"] = int(episode_lens[i])
                        stats_out.append(stats)
                        episode_returns[i] = 0.0
                        episode_lens[i] = 0
                        episode_counters[i] += 1
                        o = env.reset(seed=base_seed + 1000 * i
                                      + episode_counters[i])
                    obs_out.append(to_compact(o))
                    rewards.append(r)
                    dones.append(done)
                obs_cache = list(obs_out)
                conn.send((obs_out, np.array(rewards), np.array(dones),
                           stats_out))
            elif cmd == "preload_memo":
                # inject (model, degree) memo tables computed after fork
                # (e.g. by the batched HIP lookahead on the parent's GPU)
                lookahead_memo, init_memo = payload
                for env in envs:
                    env.lookahead_memo_preload = lookahead_memo
                    env.init_details_memo_preload = init_memo
                    cluster = getattr(env, "cluster", None)
                    if cluster is not None and hasattr(
                            cluster, "job_model_to_max_num_partitions_to_lookahead"):
                        for (model, degree), entry in (init_memo or {}).items():
                            cluster.job_model_to_max_num_partitions_to_init_details[
                                model][degree] = dict(entry)
                        for (model, degree), entry in (lookahead_memo or {}).items():
                            cluster.job_model_to_max_num_partitions_to_lookahead[
                                model][degree] = entry
                conn.send("ok")
            elif cmd == "close":
                conn.send("closed")
                return
            else:
                raise ValueError(f"unknown command {cmd}")
    except (KeyboardInterrupt, EOFError):
        pass


class SubprocVectorEnv:
    """Same interface as rollout.VectorEnv, envs sharded over processes."""

    SHM_BYTES = 16 << 20   # shared weight buffer; larger policies fall back

    def __init__(self, env_fn, num_envs: int, num_workers: Optional[int] = None,
                 base_seed: int = 0):
        if num_workers is None:
            num_workers = min(num_envs, max(1, (os.cpu_count() or 2) - 1))
        num_workers = min(num_workers, num_envs)
        self.num_envs = num_envs
        # shard envs as evenly as possible
        base = num_envs // num_workers
        rem = num_envs % num_workers
        self.shards = [base + (1 if w < rem else 0) for w in range(num_workers)]
        ctx = mp.get_context("fork")
        # fork-inherited shared buffer for the policy weights: one vectorised
        # write in the parent replaces num_workers x ~400 KB pickled
        # state_dicts per rollout
        self._shm_buf = ctx.RawArray("b", self.SHM_BYTES)
        self._shm_meta = None
        self._shm_views = None
        self.conns, self.procs = [], []
        offset = 0
        for w, count in enumerate(self.shards):
            parent, child = ctx.Pipe()
            p = ctx.Process(target=_worker_loop,
                            args=(child, count, env_fn,
                                  base_seed + 1000000 * w, self._shm_buf),
                            daemon=True)
            p.start()
            child.close()
            self.conns.append(parent)
            self.procs.append(p)
            offset += count
        self.obs: List[CompactObs] = []
        self.completed_episode_stats: List[dict] = []

    def __len__(self):
        return self.num_envs

    def reset(self):
        for conn in self.conns:
            conn.send(("reset", None))
        self.obs = []
        for conn in self.conns:
            self.obs.extend(conn.recv())
        return self.obs

    def step(self, actions: np.ndarray):
        start = 0
        for conn, count in zip(self.conns, self.shards):
            conn.send(("step", actions[start:start + count]))
            start += count
        obs, rewards, dones = [], [], []
        for conn in self.conns:
            o, r, d, stats = conn.recv()
            obs.extend(o)
            rewards.append(r)
            dones.append(d)
            self.completed_episode_stats.extend(stats)
        self.obs = obs
        return obs, np.concatenate(rewards), np.concatenate(dones)

    def rollout(self, policy, steps: int):
        """Worker-side rollout for all envs: ship CPU weights once, collect T
        steps per env locally in every worker, gather trajectories.

        Returns dict with [T, N] arrays (actions/logp/values/rewards/dones),
        obs as a flat [T*N] list (t-major), and bootstrap_values [N].
        """
        cfg = dict(policy.config)
        if self._shm_meta is None and self._shm_buf is not None:
            self._try_init_shm(policy)
        if self._shm_views is not None:
            for k, v in policy.state_dict().items():
                self._shm_views[k][...] = v.detach().cpu().numpy()
            payload = (None, steps, cfg)
        else:
            state_dict = {k: v.detach().cpu().numpy()
                          for k, v in policy.state_dict().items()}
            payload = (state_dict, steps, cfg)
        for conn in self.conns:
            conn.send(("rollout", payload))
        per_worker = []
        for conn in self.conns:
            traj, stats = conn.recv()
            per_worker.append(traj)
            self.completed_episode_stats.extend(stats)
        T = steps
        obs_flat = []
        for t in range(T):
            for traj in per_worker:
                obs_flat.extend(traj["obs"][t])
        out = {"obs": obs_flat}
        for key in ("actions", "logp", "values", "rewards", "dones", "lp_all"):
            out[key] = np.stack(
                [np.concatenate([traj[key][t] for traj in per_worker])
                 for t in range(T)])
        out["bootstrap_values"] = np.concatenate(
            [traj["bootstrap_values"] for traj in per_worker])
        return out

    def _try_init_shm(self, policy):
        """Lay out the policy's params in the fork-shared buffer and send the
        layout to every worker; falls back to pickled state_dicts when the
        policy does not fit."""
        meta, off = [], 0
        sd = {k: v.detach().cpu().numpy() for k, v in policy.state_dict().items()}
        if any(v.dtype != np.float32 for v in sd.values()):
            # a non-fp32 param would make NumpyGNNPolicy copy (not alias) the
            # shm view, silently freezing worker weights — fall back to
            # pickled state_dicts
            self._shm_buf = None
            return
        for k, v in sd.items():
            off = (off + 63) & ~63   # 64 B alignment per tensor
            meta.append((k, off, v.nbytes, tuple(v.shape), v.dtype.str))
            off += v.nbytes
        if off > len(self._shm_buf):
            self._shm_buf = None
            return
        raw = np.frombuffer(self._shm_buf, dtype=np.uint8)
        self._shm_views = {
            k: raw[o:o + nb].view(np.dtype(dt)).reshape(shape)
            for k, o, nb, shape, dt in meta}
        for conn in self.conns:
            conn.send(("shm_meta", meta))
        for conn in self.conns:
            conn.recv()
        self._shm_meta = meta

    def drain_episode_stats(self) -> List[dict]:
        out = self.completed_episode_stats
        self.completed_episode_stats = []
        return out

    def warm(self, steps: int = 96):
        """Pre-warm worker-side pipeline caches (see the ``warm`` command)."""
        for conn in self.conns:
            conn.send(("warm", steps))
        for conn in self.conns:
            conn.recv()

    def preload_memos(self, lookahead_memo, init_details_memo):
        for conn in self.conns:
            conn.send(("preload_memo", (lookahead_memo, init_details_memo)))
        for conn in self.conns:
            conn.recv()

    def close(self):
        for conn in self.conns:
            try:
                conn.send(("close", None))
                conn.recv()
            except (BrokenPipeError, EOFError):
                pass
        for p in self.procs:
            p.join(timeout=5)

    def __del__(self):
        try:
            self.close()
        except Exception:
            pass
