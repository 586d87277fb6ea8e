"""PAC-ML GNN policy, batched.

Reference: ``ddls/ml_models/models/mean_pool.py:5`` (MeanPool),
``models/gnn.py:5`` (GNN stack), ``policies/gnn_policy.py:53`` (GNNPolicy).

The reference builds one DGL graph per sample in a Python loop
(``gnn_policy.py:226-253``) — the training hot spot.  This
rebuild batches the whole minibatch into one flat edge list (global node
indices + per-node graph ids) and runs message passing with gather/segment
ops, a layout chosen so the fused HIP/MFMA kernel (ddls_amd.ops) can replace
the inner loop 1:1.

Semantics parity notes:
- message = concat(node_mlp(z_src), edge_mlp(e)); the receiving node adds a
  self-message concat(node_mlp(z_v), zeros).
- every message goes through reduce_mlp (LayerNorm -> Linear -> act) BEFORE
  the mean (mean of MLP, not MLP of mean).
- DGL's update_all zero-fills nodes with no in-edges (the self-message is only
  added for nodes that receive mail); replicated here.
"""
from __future__ import annotations

import math
from dataclasses import dataclass
from typing import Dict, List, Optional

import numpy as np
import torch
import torch.nn as nn

ACTIVATIONS = {
    "relu": nn.ReLU,
    "leaky_relu": nn.LeakyReLU,
    "tanh": nn.Tanh,
    "elu": nn.ELU,
    "sigmoid": nn.Sigmoid,
}

DEFAULT_GNN_CONFIG = {
    # tuned values from the reference (scripts/.../model/gnn.yaml:16-35)
    "in_features_node": 5,
    "in_features_edge": 2,
    "in_features_graph": 17,
    "out_features_msg": 32,
    "out_features_hidden": 64,
    "out_features_node": 16,
    "out_features_graph": 8,
    "num_rounds": 2,
    "aggregator_activation": "relu",
    "module_depth": 1,
    "fcnet_hiddens": [256],
    "fcnet_activation": "relu",
    "apply_action_mask": True,
}


@dataclass
class GraphBatch:
    """Flat batch of variable-size graphs."""
    z: torch.Tensor            # [N_total, F_node]
    e: torch.Tensor            # [E_total, F_edge]
    src: torch.Tensor          # [E_total] global node indices
    dst: torch.Tensor          # [E_total]
    graph_of_node: torch.Tensor  # [N_total] graph id per node
    num_graphs: int

    @staticmethod
    def from_padded(node_features: torch.Tensor,
                    edge_features: torch.Tensor,
                    edges_src: torch.Tensor,
                    edges_dst: torch.Tensor,
                    node_split: torch.Tensor,
                    edge_split: torch.Tensor) -> "GraphBatch":
        """Strip zero-padding and concatenate into one flat graph.

        All tensors batched [B, ...]; node_split/edge_split give the true
        counts per sample (reference obs fields of the same names).
        """
        B, Nmax = node_features.shape[0], node_features.shape[1]
        Emax = edge_features.shape[1]
        device = node_features.device
        ns = node_split.reshape(B).long()
        es = edge_split.reshape(B).long()
        node_mask = torch.arange(Nmax, device=device)[None, :] < ns[:, None]
        edge_mask = torch.arange(Emax, device=device)[None, :] < es[:, None]
        z = node_features[node_mask]
        e = edge_features[edge_mask]
        offsets = torch.cumsum(ns, 0) - ns
        src = (edges_src.long() + offsets[:, None])[edge_mask]
        dst = (edges_dst.long() + offsets[:, None])[edge_mask]
        graph_of_node = torch.repeat_interleave(
            torch.arange(B, device=device), ns)
        return GraphBatch(z=z, e=e, src=src, dst=dst,
                          graph_of_node=graph_of_node, num_graphs=B)


def _mlp(in_dim: int, out_dim: int, act, depth: int,
         input_act: bool = True) -> nn.Sequential:
    layers: List[nn.Module] = [nn.LayerNorm(in_dim), nn.Linear(in_dim, out_dim)]
    if input_act:
        layers.append(act())
    for _ in range(depth - 1):
        layers.extend([nn.Linear(out_dim, out_dim), act()])
    return nn.Sequential(*layers)


class MeanPoolLayer(nn.Module):
    """One message-passing round (reference ``mean_pool.py:5-150``)."""

    def __init__(self, in_features_node: int, in_features_edge: int,
                 out_features_msg: int, out_features_reduce: int,
                 aggregator_activation: str = "leaky_relu",
                 module_depth: int = 1):
        super().__init__()
        if module_depth < 1:
            raise ValueError("module_depth must be >= 1")
        act = ACTIVATIONS[aggregator_activation]
        half = out_features_msg // 2
        self.node_module = _mlp(in_features_node, half, act, module_depth)
        self.edge_module = _mlp(in_features_edge, half, act, module_depth)
        self.reduce_module = _mlp(out_features_msg, out_features_reduce, act,
                                  module_depth)
        self.out_features_msg = out_features_msg
        self.out_features_reduce = out_features_reduce

    def forward(self, z: torch.Tensor, e: torch.Tensor, src: torch.Tensor,
                dst: torch.Tensor) -> torch.Tensor:
        N = z.shape[0]
        hn = self.node_module(z)                       # [N, msg/2]
        he = self.edge_module(e)                       # [E, msg/2]
        msg_edge = torch.cat([hn[src], he], dim=-1)    # [E, msg]
        msg_self = torch.cat([hn, torch.zeros_like(hn)], dim=-1)  # [N, msg]
        r_edge = self.reduce_module(msg_edge)          # [E, out]
        r_self = self.reduce_module(msg_self)          # [N, out]
        out = torch.zeros(N, r_self.shape[-1], dtype=r_edge.dtype,
                          device=z.device)
        out.index_add_(0, dst, r_edge)
        in_deg = torch.zeros(N, dtype=z.dtype, device=z.device)
        in_deg.index_add_(0, dst, torch.ones_like(dst, dtype=z.dtype))
        has_mail = in_deg > 0
        total = out + r_self
        count = (in_deg + 1).unsqueeze(-1)
        mean = total / count
        # DGL zero-fills nodes with no incoming messages
        return torch.where(has_mail.unsqueeze(-1), mean,
                           torch.zeros_like(mean))


class GNN(nn.Module):
    """Stack of >= 2 MeanPool rounds (reference ``models/gnn.py:5-89``)."""

    def __init__(self, config: Dict):
        super().__init__()
        if config["num_rounds"] < 2:
            raise ValueError("num_rounds must be >= 2")
        layers = [MeanPoolLayer(config["in_features_node"],
                                config["in_features_edge"],
                                config["out_features_msg"],
                                config["out_features_hidden"],
                                config["aggregator_activation"],
                                config["module_depth"])]
        for _ in range(config["num_rounds"] - 2):
            layers.append(MeanPoolLayer(config["out_features_hidden"],
                                        config["in_features_edge"],
                                        config["out_features_msg"],
                                        config["out_features_hidden"],
                                        config["aggregator_activation"],
                                        config["module_depth"]))
        layers.append(MeanPoolLayer(config["out_features_hidden"],
                                    config["in_features_edge"],
                                    config["out_features_msg"],
                                    config["out_features_node"],
                                    config["aggregator_activation"],
                                    config["module_depth"]))
        self.layers = nn.ModuleList(layers)

    def forward(self, batch: GraphBatch) -> torch.Tensor:
        z = batch.z
        for layer in self.layers:
            z = layer(z, batch.e, batch.src, batch.dst)
        return z


class GNNPolicy(nn.Module):
    """GNN -> per-graph mean node embedding ++ graph-feature embedding ->
    policy logits + value (reference ``gnn_policy.py:53-276``; the FC head
    mirrors RLlib's FullyConnectedNetwork with a separate value branch)."""

    def __init__(self, num_actions: int, config: Optional[Dict] = None):
        super().__init__()
        cfg = dict(DEFAULT_GNN_CONFIG)
        if config:
            cfg.update(config)
        self.config = cfg
        self.num_actions = num_actions
        self.gnn = GNN(cfg)

        act = ACTIVATIONS[cfg["aggregator_activation"]]
        # graph module input = graph feats + action mask (appended by the obs fn)
        in_graph = cfg["in_features_graph"] + num_actions
        graph_layers: List[nn.Module] = [nn.LayerNorm(in_graph),
                                         nn.Linear(in_graph, cfg["out_features_graph"])]
        for _ in range(cfg["module_depth"] - 1):
            graph_layers.extend([nn.Linear(cfg["out_features_graph"],
                                           cfg["out_features_graph"]), act()])
        self.graph_module = nn.Sequential(*graph_layers)

        emb_dim = cfg["out_features_graph"] + cfg["out_features_node"]
        fc_act = ACTIVATIONS[cfg["fcnet_activation"]]
        hiddens = cfg["fcnet_hiddens"]

        def branch(out_dim):
            layers, d = [], emb_dim
            for h in hiddens:
                layers.extend([nn.Linear(d, h), fc_act()])
                d = h
            layers.append(nn.Linear(d, out_dim))
            return nn.Sequential(*layers)

        self.policy_branch = branch(num_actions)
        self.value_branch = branch(1)

    def forward(self, obs: Dict[str, torch.Tensor]):
        """obs: batched padded observation dict (torch tensors, [B, ...])."""
        batch = GraphBatch.from_padded(
            obs["node_features"], obs["edge_features"], obs["edges_src"],
            obs["edges_dst"], obs["node_split"], obs["edge_split"])
        node_emb = self.gnn(batch)                      # [N_total, out_node]
        # per-graph mean of node embeddings
        B = batch.num_graphs
        sums = torch.zeros(B, node_emb.shape[-1], dtype=node_emb.dtype,
                           device=node_emb.device)
        sums.index_add_(0, batch.graph_of_node, node_emb)
        counts = torch.zeros(B, dtype=node_emb.dtype, device=node_emb.device)
        counts.index_add_(0, batch.graph_of_node,
                          torch.ones_like(batch.graph_of_node, dtype=node_emb.dtype))
        graph_node_emb = sums / counts.clamp(min=1).unsqueeze(-1)

        graph_emb = self.graph_module(obs["graph_features"])
        final_emb = torch.cat([graph_node_emb, graph_emb], dim=-1)

        logits = self.policy_branch(final_emb)
        value = self.value_branch(final_emb).squeeze(-1)

        if self.config["apply_action_mask"]:
            mask = obs["action_mask"].to(logits.dtype)
            inf_mask = torch.clamp(torch.log(mask),
                                   min=torch.finfo(torch.float32).min)
            logits = logits + inf_mask
        return logits, value
