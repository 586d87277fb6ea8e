"""Numpy inference mirror of GNNPolicy for env-worker rollouts.

The policy is ~2e4 params and rollout graphs are ~60 nodes; a torch forward
on such shapes is dominated by op-dispatch overhead (~15 ms on one CPU
thread).  This mirror runs the identical computation in numpy (~0.5 ms),
used only for worker-side action sampling; the SGD update always recomputes
through torch autograd (tiny fp differences only shift the importance ratio
by rounding noise).
"""
from __future__ import annotations

from typing import Dict, Tuple

import numpy as np

from ..rl.rollout import CompactObs

LN_EPS = 1e-5


def _ln(x, g, b):
    # single-pass, raw ufunc reductions (np.mean's wrapper overhead shows up
    # at 9 calls per env step)
    inv_k = 1.0 / x.shape[-1]
    mu = x.sum(axis=-1, keepdims=True)
    mu *= inv_k
    d = x - mu
    var = np.einsum("...k,...k->...", d, d)
    inv = 1.0 / np.sqrt(var * inv_k + LN_EPS)
    d *= inv[..., None]
    d *= g
    d += b
    return d


def _relu(x):
    return np.maximum(x, 0.0)


class _MLP:
    """LayerNorm -> Linear [-> act] [-> (Linear -> act) * (depth-1)]."""

    def __init__(self, params: Dict[str, np.ndarray], prefix: str,
                 input_act: bool = True):
        self.ln_g = params[f"{prefix}.0.weight"]
        self.ln_b = params[f"{prefix}.0.bias"]
        self.W = params[f"{prefix}.1.weight"]
        self.b = params[f"{prefix}.1.bias"]
        self.input_act = input_act
        # depth-1 extra layers
        self.extra = []
        i = 3 if input_act else 2
        while f"{prefix}.{i}.weight" in params:
            self.extra.append((params[f"{prefix}.{i}.weight"],
                               params[f"{prefix}.{i}.bias"]))
            i += 2

    def __call__(self, x):
        y = _ln(x, self.ln_g, self.ln_b) @ self.W.T + self.b
        if self.input_act:
            y = _relu(y)
        for W, b in self.extra:
            y = _relu(y @ W.T + b)
        return y


class _Branch:
    """Linear->act stack + final Linear (policy/value branches)."""

    def __init__(self, params, prefix):
        self.layers = []
        i = 0
        while f"{prefix}.{i}.weight" in params:
            self.layers.append((params[f"{prefix}.{i}.weight"],
                                params[f"{prefix}.{i}.bias"]))
            i += 2
        # find trailing layer index (last Linear has no activation after it)

    def __call__(self, x):
        for k, (W, b) in enumerate(self.layers):
            x = x @ W.T + b
            if k < len(self.layers) - 1:
                x = _relu(x)
        return x


class NumpyGNNPolicy:
    """Inference-only mirror of models.gnn.GNNPolicy."""

    def __init__(self, state_dict: Dict[str, np.ndarray], config: Dict,
                 num_actions: int):
        # NB: np.asarray must be a VIEW for the shared-memory weight path
        # (subproc_env fork-shared buffer) to see live weight updates; a
        # non-fp32 or non-contiguous param would silently copy,
        # so assert aliasing and let subproc_env fall back to pickling.
        p = {k: np.asarray(v, dtype=np.float32) for k, v in state_dict.items()}
        self.aliases_inputs = all(
            isinstance(v, np.ndarray) and (p[k] is v or p[k].base is v)
            for k, v in state_dict.items())
        self.config = config
        self.num_actions = num_actions
        n_layers = config["num_rounds"]
        self.layers = []
        for li in range(n_layers):
            self.layers.append({
                "node": _MLP(p, f"gnn.layers.{li}.node_module"),
                "edge": _MLP(p, f"gnn.layers.{li}.edge_module"),
                "reduce": _MLP(p, f"gnn.layers.{li}.reduce_module"),
            })
        self.graph_module = _MLP(p, "graph_module", input_act=False)
        self.policy_branch = _Branch(p, "policy_branch")
        self.value_branch = _Branch(p, "value_branch")

    @staticmethod
    def _segments(obs):
        """Per-obs CSR segments for the reduceat-based scatter (np.add.at is
        an order of magnitude slower); cached on the CompactObs, which the
        worker loop memoises per underlying env obs dict."""
        seg = getattr(obs, "_np_seg", None)
        if seg is None:
            dst = obs.edges_dst
            n = len(obs.node_features)
            order = np.argsort(dst, kind="stable")
            dst_sorted = dst[order]
            uniq, starts = np.unique(dst_sorted, return_index=True)
            in_deg = np.bincount(dst, minlength=n).astype(np.float32)
            seg = (order, starts, uniq, in_deg)
            try:
                obs._np_seg = seg
            except Exception:
                pass
        return seg

    def _meanpool(self, layer, z, e, src, dst, seg):
        hn = layer["node"](z)
        he = layer["edge"](e)
        msg_edge = np.concatenate([hn[src], he], axis=-1)
        msg_self = np.concatenate([hn, np.zeros_like(hn)], axis=-1)
        r_edge = layer["reduce"](msg_edge)
        r_self = layer["reduce"](msg_self)
        N = z.shape[0]
        order, starts, uniq, in_deg = seg
        out = np.zeros((N, r_self.shape[-1]), dtype=np.float32)
        if len(order):
            out[uniq] = np.add.reduceat(r_edge[order], starts, axis=0)
        total = out + r_self
        mean = total / (in_deg + 1.0)[:, None]
        mean[in_deg == 0] = 0.0
        return mean

    def forward(self, obs: CompactObs) -> Tuple[np.ndarray, float]:
        z = obs.node_features
        e = obs.edge_features
        src, dst = obs.edges_src, obs.edges_dst
        seg = self._segments(obs)
        for layer in self.layers:
            z = self._meanpool(layer, z, e, src, dst, seg)
        graph_node_emb = z.mean(axis=0) if len(z) else np.zeros(
            self.config["out_features_node"], dtype=np.float32)
        graph_emb = self.graph_module(obs.graph_features[None, :])[0]
        final = np.concatenate([graph_node_emb, graph_emb])
        logits = self.policy_branch(final[None, :])[0]
        value = float(self.value_branch(final[None, :])[0, 0])
        if self.config.get("apply_action_mask", True):
            with np.errstate(divide="ignore"):
                inf_mask = np.maximum(np.log(obs.action_mask),
                                      np.finfo(np.float32).min)
            logits = logits + inf_mask
        return logits.astype(np.float32), value

    def act(self, obs: CompactObs, rng: np.random.RandomState
            ) -> Tuple[int, float, float]:
        """Sample a masked action; returns (action, logp, value)."""
        a, logp, value, _ = self.act_full(obs, rng)
        return a, logp, value

    def act_full(self, obs: CompactObs, rng: np.random.RandomState
                 ) -> Tuple[int, float, float, np.ndarray]:
        """act() plus the full behavior log-prob vector (for the analytic
        KL(old || new) used by the adaptive KL coefficient)."""
        logits, value = self.forward(obs)
        x = logits - logits.max()
        p = np.exp(x)
        Z = p.sum()
        p /= Z
        lp_all = (x - np.log(Z)).astype(np.float32)
        a = int(rng.choice(len(p), p=p))
        logp = float(np.log(max(p[a], 1e-45)))
        return a, logp, value, lp_all
