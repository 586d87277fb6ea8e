"""GPU-resident vectorised env for PPO rollouts.

Replaces the CPU subprocess rollout workers (``rl/subproc_env.py``) on GPU
machines: the env step runs as ONE batched HIP kernel over all envs
(``cluster/gpu_engine.py``) and the policy forward collapses to a per-model
GNN embedding cache + a tiny per-step head — exact, because within one
rollout the weights are fixed and the GNN sees only static per-model
node/edge features; the only per-step inputs are the 17+|A| graph features
and the mask (``envs/observation.py``).

The SGD update path is unchanged: rollouts return the same trajectory dict
as SubprocVectorEnv (obs as CompactObs whose node/edge arrays are shared
per-model views — zero copies).
"""
from __future__ import annotations

import sys
import time
from typing import Dict, List

import numpy as np
import torch

from ..cluster.gpu_engine import GpuEngine
from ..cluster.vec_engine import (compile_engine_spec, drain_episode_schedule)
from ..models.gnn import GraphBatch, graph_mean
from .rollout import CompactObs


class EngineVectorEnv:
    """Same trainer-facing API as SubprocVectorEnv, GPU-resident."""

    def __init__(self, env_fn, num_envs: int, device, base_seed: int = 0,
                 spec=None, proto_env=None, verbose: bool = False):
        self.device = torch.device(device)
        self.num_envs = num_envs
        self.base_seed = base_seed
        t0 = time.perf_counter()
        if proto_env is None:
            proto_env = env_fn()
            proto_env.reset(seed=987654321)
        self.proto_env = proto_env
        if spec is None:
            spec = compile_engine_spec(
                proto_env, lookahead_device=self.device
                if torch.cuda.is_available() else None)
        self.spec = spec
        self.gen = proto_env.cluster.jobs_generator
        self.episode_counters = [0] * num_envs
        self.scheds = [self._drain(i) for i in range(num_envs)]
        cap = max(s.n for s in self.scheds) + 8
        if self.device.type == "cuda":
            self.eng = GpuEngine(spec, B=num_envs, device=self.device,
                                 n_jobs_cap=cap)
        else:
            # CPU backend: the parity-tested mirror as a vectorised engine
            # (engine-speed rollouts without subprocess env workers)
            from ..cluster.vec_engine import CpuEngine
            self.eng = CpuEngine(spec, B=num_envs, n_jobs_cap=cap)
        for b in range(num_envs):
            self.eng.reset_env(b, self.scheds[b])
        self.completed_episode_stats: List[dict] = []
        # per-model static tensors for collate-at-update + embedding cache
        self._model_nf = [ms.node_features for ms in spec.models]
        self._model_ef = [ms.edge_features for ms in spec.models]
        self._model_src = [ms.edges_src for ms in spec.models]
        self._model_dst = [ms.edges_dst for ms in spec.models]
        self._models_batch = self._build_models_batch()
        # running completed-jobs JCT accounting (for bench mid-episode JCT)
        self._jct_base_count = 0
        self._jct_base_sum = 0.0
        if verbose:
            # stderr: bench.py's stdout must stay a single JSON line
            print(f"[engine_env] init {num_envs} envs in "
                  f"{time.perf_counter() - t0:.2f}s", file=sys.stderr)

    # ------------------------------------------------------------------
    def _drain(self, b: int):
        seed = self.base_seed + 1000 * b + self.episode_counters[b]
        return drain_episode_schedule(self.gen, self.spec, seed=seed)

    def _build_models_batch(self) -> GraphBatch:
        """One flat GraphBatch holding every model's (static) graph."""
        dev = self.device
        z = np.concatenate(self._model_nf)
        e = np.concatenate(self._model_ef)
        ns = np.array([nf.shape[0] for nf in self._model_nf], dtype=np.int64)
        offs = np.concatenate([[0], np.cumsum(ns)[:-1]])
        src = np.concatenate([s + o for s, o in zip(self._model_src, offs)])
        dst = np.concatenate([d + o for d, o in zip(self._model_dst, offs)])
        gid = np.repeat(np.arange(len(ns), dtype=np.int64), ns)
        t = lambda a, dt: torch.as_tensor(a, device=dev, dtype=dt)
        return GraphBatch(z=t(z, torch.float32), e=t(e, torch.float32),
                          src=t(src, torch.int64), dst=t(dst, torch.int64),
                          graph_of_node=t(gid, torch.int64),
                          num_graphs=len(ns))

    def __len__(self):
        return self.num_envs

    # ------------------------------------------------------------------
    def _obs_list(self) -> List[CompactObs]:
        gf = self.eng.T["obs_gf"].cpu().numpy()
        mask = self.eng.T["obs_mask"].cpu().numpy()
        mids = self.eng.T["obs_model"].cpu().numpy()
        return [self._compact(int(mids[b]),
                              np.concatenate([gf[b], mask[b]]), mask[b])
                for b in range(self.num_envs)]

    def _compact(self, mid: int, gfull: np.ndarray,
                 mask: np.ndarray) -> CompactObs:
        return CompactObs(node_features=self._model_nf[mid],
                          edge_features=self._model_ef[mid],
                          edges_src=self._model_src[mid],
                          edges_dst=self._model_dst[mid],
                          graph_features=gfull, action_mask=mask,
                          model_id=mid)

    def reset(self) -> List[CompactObs]:
        return self._obs_list()

    # ------------------------------------------------------------------
    @torch.no_grad()
    def _head_forward(self, policy, model_emb, obs_model, gfull, mask):
        """Mirror of GNNPolicy.forward_flat with the GNN replaced by the
        cached per-model embeddings."""
        graph_emb = policy.graph_module(gfull)
        final = torch.cat([model_emb[obs_model], graph_emb], dim=-1)
        logits = policy.policy_branch(final)
        value = policy.value_branch(final).squeeze(-1)
        if policy.config["apply_action_mask"]:
            inf_mask = torch.clamp(torch.log(mask),
                                   min=torch.finfo(torch.float32).min)
            logits = logits + inf_mask
        return logits, value

    @torch.no_grad()
    def rollout(self, policy, steps: int) -> Dict[str, np.ndarray]:
        dev = self.device
        B = self.num_envs
        A = self.spec.A
        eng = self.eng
        # per-model GNN embedding cache (weights fixed within the rollout)
        node_emb = policy.gnn(self._models_batch)
        model_emb = graph_mean(node_emb, self._models_batch)   # [M, out_node]

        T = steps
        buf_gfull = torch.empty((T, B, 17 + A), device=dev)
        buf_model = torch.empty((T, B), dtype=torch.int64, device=dev)
        buf_actions = torch.empty((T, B), dtype=torch.int64, device=dev)
        buf_lp_all = torch.empty((T, B, A), device=dev)
        buf_logp = torch.empty((T, B), device=dev)
        buf_values = torch.empty((T, B), device=dev)
        buf_rewards = torch.empty((T, B), device=dev)
        buf_dones = torch.empty((T, B), dtype=torch.bool, device=dev)

        for t in range(T):
            gf = eng.T["obs_gf"]
            mask = eng.T["obs_mask"]
            gfull = torch.cat([gf, mask], dim=1)
            obs_model = eng.T["obs_model"].long()
            logits, value = self._head_forward(policy, model_emb, obs_model,
                                               gfull, mask)
            dist = torch.distributions.Categorical(logits=logits)
            actions = dist.sample()
            buf_gfull[t] = gfull
            buf_model[t] = obs_model
            buf_actions[t] = actions
            buf_lp_all[t] = dist.logits   # normalized log-probs (Categorical)
            buf_logp[t] = dist.log_prob(actions)
            buf_values[t] = value
            eng.step(actions)          # syncs status internally
            buf_rewards[t] = eng.T["reward"].float()
            done = eng.T["done"] != 0
            buf_dones[t] = done
            if bool(done.any()):
                for b in torch.nonzero(done).flatten().tolist():
                    es = eng.episode_stats(b)
                    self.completed_episode_stats.append(es)
                    self._jct_base_count += es["num_jobs_completed"]
                    self._jct_base_sum += float(
                        np.sum(es["job_completion_time"]))
                    self.episode_counters[b] += 1
                    self.scheds[b] = self._drain(b)
                    eng.reset_env(b, self.scheds[b])

        # bootstrap values of the final obs
        gf = eng.T["obs_gf"]
        mask = eng.T["obs_mask"]
        gfull = torch.cat([gf, mask], dim=1)
        _, bootstrap = self._head_forward(policy, model_emb,
                                          eng.T["obs_model"].long(), gfull,
                                          mask)

        # D2H once; build CompactObs over shared per-model arrays
        gfull_np = buf_gfull.cpu().numpy()
        model_np = buf_model.cpu().numpy()
        obs_flat = [self._compact(int(model_np[t, b]), gfull_np[t, b],
                                  gfull_np[t, b, 17:])
                    for t in range(T) for b in range(B)]
        return {
            "obs": obs_flat,
            "actions": buf_actions.cpu().numpy(),
            "lp_all": buf_lp_all.cpu().numpy(),
            "logp": buf_logp.cpu().numpy(),
            "values": buf_values.cpu().numpy(),
            "rewards": buf_rewards.cpu().numpy(),
            "dones": buf_dones.cpu().numpy(),
            "bootstrap_values": bootstrap.cpu().numpy(),
        }

    # ------------------------------------------------------------------
    def drain_episode_stats(self) -> List[dict]:
        out = self.completed_episode_stats
        self.completed_episode_stats = []
        return out

    def jct_running_stats(self):
        """(count, sum) of completed-job JCTs across all envs, INCLUDING jobs
        completed inside still-running episodes (for mid-episode bench JCT
        reporting)."""
        from ..cluster.vec_engine import COMPLETED
        count = self._jct_base_count
        total = self._jct_base_sum
        status = self.eng.T["log_status"].cpu().numpy()
        t_arr = self.eng.T["log_t_arr"].cpu().numpy()
        t_end = self.eng.T["log_t_end"].cpu().numpy()
        comp = status == COMPLETED
        count += int(comp.sum())
        total += float((t_end[comp] - t_arr[comp]).sum())
        return count, total

    def close(self):
        pass
