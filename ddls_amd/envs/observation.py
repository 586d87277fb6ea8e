"""PAC-ML partitioning observation.

Reference: ``ddls/environments/ramp_job_partitioning/observations/
ramp_job_partitioning_observation.py:15`` — Dict obs of {action_set,
action_mask, node_features (5/node), edge_features (2/edge), graph_features
(17 + action mask appended), edges_src, edges_dst, node_split, edge_split},
zero-padded to max_nodes / fully-connected max_edges, per-feature min-max
normalisation from the JobsGenerator's jobs_params.
"""
from __future__ import annotations

from typing import Optional

import numpy as np

from ..agents.placement_utils import get_block, get_block_shapes, get_factor_pairs
from . import spaces
from .base import DDLSObservationFunction


class RampJobPartitioningObservation(DDLSObservationFunction):
    def __init__(self,
                 max_partitions_per_op: int,
                 pad_obs_kwargs: Optional[dict] = None,
                 machine_epsilon: float = 1e-7):
        self.max_partitions_per_op = max_partitions_per_op
        self.pad_obs_kwargs = pad_obs_kwargs
        self.machine_epsilon = machine_epsilon
        self.observation_space = None
        self.node_features_low, self.node_features_high = 0, 1
        self.edge_features_low, self.edge_features_high = 0, 1
        self.graph_features_low, self.graph_features_high = 0, 1

    def reset(self, env):
        if self.pad_obs_kwargs is not None:
            self.max_nodes = self.pad_obs_kwargs["max_nodes"]
            self.max_edges = int(self.max_nodes * (self.max_nodes - 1) / 2)
        else:
            self.max_nodes, self.max_edges = 0, 0
        obs = self.extract(env, done=False)
        self.observation_space = spaces.Dict({
            "action_set": spaces.Box(low=int(obs["action_set"].min()),
                                     high=int(obs["action_set"].max()),
                                     shape=obs["action_set"].shape,
                                     dtype=obs["action_set"].dtype),
            "action_mask": spaces.Box(low=0, high=1, shape=obs["action_mask"].shape,
                                      dtype=obs["action_mask"].dtype),
            "node_features": spaces.Box(low=0, high=1, shape=obs["node_features"].shape,
                                        dtype=obs["node_features"].dtype),
            "edge_features": spaces.Box(low=0, high=1, shape=obs["edge_features"].shape,
                                        dtype=obs["edge_features"].dtype),
            "graph_features": spaces.Box(low=0, high=1, shape=obs["graph_features"].shape,
                                         dtype=obs["graph_features"].dtype),
            "edges_src": spaces.Box(low=0, high=self.max_nodes - 1,
                                    shape=obs["edges_src"].shape,
                                    dtype=obs["edges_src"].dtype),
            "edges_dst": spaces.Box(low=0, high=self.max_nodes - 1,
                                    shape=obs["edges_dst"].shape,
                                    dtype=obs["edges_dst"].dtype),
            "node_split": spaces.Box(low=0, high=self.max_nodes, shape=(1,),
                                     dtype=obs["node_split"].dtype),
            "edge_split": spaces.Box(low=0, high=self.max_edges, shape=(1,),
                                     dtype=obs["edge_split"].dtype),
        })
        return obs

    # ------------------------------------------------------------------
    def get_action_set_and_action_mask(self, env):
        """Reference ``:80-131``: action 0 always valid; even degrees with a
        feasible RAMP block shape and enough free workers are valid."""
        ramp_shape = env.cluster.topology.shape
        action_set, action_mask = [0], [True]
        num_avail = (env.cluster.topology.num_workers
                     - len(env.cluster.mounted_workers))
        for action in range(1, env.max_partitions_per_op + 1):
            action_set.append(action)
            is_valid = False
            if (action > 1 and action % 2 == 0) or action == 1:
                if action <= env.max_partitions_per_op and action <= num_avail:
                    if action == 1:
                        is_valid = True
                    else:
                        pairs = get_factor_pairs(action)
                        block_shapes = get_block_shapes(pairs, ramp_shape)
                        b = []
                        for shape in block_shapes:
                            b.extend(get_block(shape[0], shape[1], shape[2],
                                               ramp_shape))
                        is_valid = len(b) > 0
            action_mask.append(is_valid)
        return action_set, action_mask

    # ------------------------------------------------------------------
    def extract(self, env, done: bool):
        job = self._get_job_to_encode(env)
        return self._encode_obs(job, env)

    def _get_job_to_encode(self, env):
        return next(iter(env.cluster.job_queue.jobs.values()))

    def _encode_obs(self, job, env):
        g = job.graph
        if self.pad_obs_kwargs is not None:
            if g.n > self.max_nodes:
                raise ValueError(f"job has {g.n} nodes > max_nodes {self.max_nodes}")
            if g.m > self.max_edges:
                raise ValueError(f"job has {g.m} edges > max_edges {self.max_edges}")

        action_set, action_mask = self.get_action_set_and_action_mask(env)
        obs = {
            "action_set": np.array(action_set, dtype=np.int16),
            "action_mask": np.array(action_mask, dtype=np.int16),
            "node_features": self._node_features(job, env.cluster),
            "edge_features": self._edge_features(job, env.cluster),
            "graph_features": self._graph_features(job, env.cluster),
            "edges_src": g.src.astype(np.float32),
            "edges_dst": g.dst.astype(np.float32),
            "node_split": np.array([np.nan], dtype=np.float32),
            "edge_split": np.array([np.nan], dtype=np.float32),
        }
        obs["graph_features"] = np.concatenate(
            [obs["graph_features"], obs["action_mask"].astype(np.float32)])
        if self.pad_obs_kwargs is not None:
            obs = self._pad_obs(obs)
        for key, val in obs.items():
            if key not in ("node_split", "edge_split"):
                if not np.isfinite(val).all():
                    raise ValueError(f"{key} in observation contains NaN/inf")
        return obs

    def _pad_obs(self, obs):
        nf, ef = obs["node_features"], obs["edge_features"]
        n, m = len(nf), len(ef)
        out = dict(obs)
        out["node_features"] = np.concatenate(
            [nf, np.zeros((self.max_nodes - n, nf.shape[1]), dtype=np.float32)])
        out["edge_features"] = np.concatenate(
            [ef, np.zeros((self.max_edges - m, ef.shape[1]), dtype=np.float32)])
        out["edges_src"] = np.concatenate(
            [obs["edges_src"], np.zeros(self.max_edges - m, dtype=np.float32)])
        out["edges_dst"] = np.concatenate(
            [obs["edges_dst"], np.zeros(self.max_edges - m, dtype=np.float32)])
        out["node_split"] = np.array([n], dtype=np.float32)
        out["edge_split"] = np.array([m], dtype=np.float32)
        return out

    # ------------------------------------------------------------------
    def _node_features(self, job, cluster) -> np.ndarray:
        """5 features/node (reference ``_get_op_features:522-621``):
        [compute_cost_norm (per device type), is_highest_compute,
        memory_cost_norm, is_highest_memory, node_depth_norm]."""
        g = job.graph
        d = job.details
        eps = self.machine_epsilon
        cols = []
        for dt in sorted(cluster.topology.worker_types):
            mx = d["max_compute_cost"][dt]
            cc = g.compute_cost[dt] / mx if mx != 0 else np.zeros(g.n)
            cols.append(cc)
            is_max_c = np.zeros(g.n)
            is_max_c[d["max_compute_node"][dt]] = 1.0
            cols.append(is_max_c)
        mc = (g.memory_cost / d["max_memory_cost"]
              if d["max_memory_cost"] != 0 else np.zeros(g.n))
        cols.append(mc)
        is_max_m = np.zeros(g.n)
        is_max_m[d["max_memory_node"]] = 1.0
        cols.append(is_max_m)
        cols.append(d["node_to_depth"] / d["max_depth"] if d["max_depth"] else
                    np.zeros(g.n))
        feats = np.stack(cols, axis=1).astype(np.float32)
        feats[feats < self.node_features_low] += eps
        if feats.min() < self.node_features_low or feats.max() > self.node_features_high:
            raise ValueError("node features out of [0, 1]")
        return feats

    def _edge_features(self, job, cluster) -> np.ndarray:
        """2 features/edge: [size_norm, is_highest_size]
        (reference ``_get_dep_features:503-520``)."""
        g = job.graph
        d = job.details
        size = (g.size / d["max_dep_size"] if d["max_dep_size"] != 0
                else np.zeros(g.m))
        is_max = np.zeros(g.m)
        if d["max_dep_size_dep"] is not None:
            is_max[d["max_dep_size_dep"]] = 1.0
        feats = np.stack([size, is_max], axis=1).astype(np.float32)
        feats[feats < self.edge_features_low] += self.machine_epsilon
        if feats.min() < 0 or feats.max() > 1:
            raise ValueError("edge features out of [0, 1]")
        return feats

    def _graph_features(self, job, cluster) -> np.ndarray:
        """15 job features + 2 network features = 17
        (reference ``_get_job_features:358-446`` +
        ``_get_network_graph_features:475-498``)."""
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
            norm(d["max_acceptable_job_completion_time"][dt],
                 p["min_max_acceptable_job_completion_times"],
                 p["max_max_acceptable_job_completion_times"]),
            norm(job.max_acceptable_job_completion_time_frac,
                 p["min_max_acceptable_job_completion_time_fracs"],
                 p["max_max_acceptable_job_completion_time_fracs"]),
            job.max_acceptable_job_completion_time_frac,
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
        # network graph features
        feats.append(len(cluster.mounted_workers) / cluster.topology.num_workers)
        feats.append(len(cluster.jobs_running) / cluster.topology.num_workers)

        feats = np.array(feats, dtype=np.float32)
        feats[feats < self.graph_features_low] += self.machine_epsilon
        if feats.min() < 0 or feats.max() > 1:
            raise ValueError(f"graph features out of [0, 1]: {feats}")
        return feats
