"""Vectorised env stepping + compact observation collation.

The reference collects rollouts via Ray actor RPC (RLlib workers, SURVEY.md
K6/K8); this rebuild steps a batch of in-process envs and batches ALL policy
forwards across envs into one GNN pass (no per-sample graph loop, no padding
in the buffer: observations are stored compact and flattened straight into a
GraphBatch).
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import Dict, List

import numpy as np
import torch

from ..models.gnn import GraphBatch


@dataclass
class CompactObs:
    """Unpadded single observation (numpy)."""
    node_features: np.ndarray   # [n, F]
    edge_features: np.ndarray   # [m, Fe]
    edges_src: np.ndarray       # [m]
    edges_dst: np.ndarray       # [m]
    graph_features: np.ndarray  # [G]
    action_mask: np.ndarray     # [A]
    # workload-model id when the obs comes from the vectorised engine (the
    # node/edge arrays are then shared per-model views); -1 = unknown.
    # Enables the cached-models SGD path (graph_step.py): GNN once per
    # distinct model per minibatch, gradients summed by index_select.
    model_id: int = -1

    @staticmethod
    def from_obs(obs: Dict[str, np.ndarray]) -> "CompactObs":
        ns, es = obs.get("node_split"), obs.get("edge_split")
        if ns is not None and np.isfinite(ns).all():
            n, m = int(ns[0]), int(es[0])
        else:
            n, m = len(obs["node_features"]), len(obs["edge_features"])
        return CompactObs(
            node_features=np.asarray(obs["node_features"][:n], dtype=np.float32),
            edge_features=np.asarray(obs["edge_features"][:m], dtype=np.float32),
            edges_src=np.asarray(obs["edges_src"][:m], dtype=np.int64),
            edges_dst=np.asarray(obs["edges_dst"][:m], dtype=np.int64),
            graph_features=np.asarray(obs["graph_features"], dtype=np.float32),
            action_mask=np.asarray(obs["action_mask"], dtype=np.float32))


def collate(obs_list: List[CompactObs], device) -> Dict[str, torch.Tensor]:
    """Build the flat GraphBatch inputs + stacked graph features/masks."""
    ns = np.array([len(o.node_features) for o in obs_list], dtype=np.int64)
    offsets = np.concatenate([[0], np.cumsum(ns)[:-1]])
    z = np.concatenate([o.node_features for o in obs_list])
    e = np.concatenate([o.edge_features for o in obs_list])
    src = np.concatenate([o.edges_src + off for o, off in zip(obs_list, offsets)])
    dst = np.concatenate([o.edges_dst + off for o, off in zip(obs_list, offsets)])
    graph_of_node = np.repeat(np.arange(len(obs_list), dtype=np.int64), ns)
    gf = np.stack([o.graph_features for o in obs_list])
    mask = np.stack([o.action_mask for o in obs_list])
    t = lambda a, dt: torch.as_tensor(a, device=device, dtype=dt)
    return {
        "batch": GraphBatch(z=t(z, torch.float32), e=t(e, torch.float32),
                            src=t(src, torch.int64), dst=t(dst, torch.int64),
                            graph_of_node=t(graph_of_node, torch.int64),
                            num_graphs=len(obs_list)),
        "graph_features": t(gf, torch.float32),
        "action_mask": t(mask, torch.float32),
    }


class VectorEnv:
    """Steps N independent env copies; auto-resets finished envs."""

    def __init__(self, env_fns: List, base_seed: int = 0):
        self.envs = [fn() for fn in env_fns]
        self.base_seed = base_seed
        self.episode_counters = [0] * len(self.envs)
        self.obs: List[CompactObs] = []
        self.episode_returns = np.zeros(len(self.envs))
        self.episode_lens = np.zeros(len(self.envs), dtype=np.int64)
        self.completed_episode_stats: List[dict] = []

    def __len__(self):
        return len(self.envs)

    def reset(self):
        self.obs = []
        for i, env in enumerate(self.envs):
            o = env.reset(seed=self.base_seed + 1000 * i)
            self.obs.append(CompactObs.from_obs(o))
        return self.obs

    def step(self, actions: np.ndarray):
        rewards = np.zeros(len(self.envs), dtype=np.float64)
        dones = np.zeros(len(self.envs), dtype=bool)
        for i, env in enumerate(self.envs):
            o, r, done, _info = env.step(int(actions[i]))
            rewards[i] = r
            dones[i] = done
            self.episode_returns[i] += r
            self.episode_lens[i] += 1
            if done:
                stats = dict(env.cluster.episode_stats)
                stats["episode_return"] = float(self.episode_returns[i])
                stats["episode_len"] = int(self.episode_lens[i])
                self.completed_episode_stats.append(stats)
                self.episode_returns[i] = 0.0
                self.episode_lens[i] = 0
                self.episode_counters[i] += 1
                o = env.reset(seed=self.base_seed + 1000 * i
                              + self.episode_counters[i])
            self.obs[i] = CompactObs.from_obs(o)
        return self.obs, rewards, dones

    def drain_episode_stats(self) -> List[dict]:
        out = self.completed_episode_stats
        self.completed_episode_stats = []
        return out
