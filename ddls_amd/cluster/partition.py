"""Graph partition transforms: data-parallel clone + intra-op (model) split.

Reference: ``ddls/environments/ramp_cluster/agents/partitioners/utils.py``
(``data_split_node:5``, ``model_split_node:42``).  Behavioural parity notes:

- ``data_split`` overwrites every edge size with the SOURCE node's memory_cost.
- ``model_split`` processes split ops in the given order; in-edge sizes are
  parent_memory/n_splits, out-edge sizes are DESTINATION_memory/n_splits (the
  reference formula), in-features applied before out-features so out wins on
  conflicts; backward sub-ops get all-to-all bidirectional sync edges sized by
  the split op's (already divided) memory cost — the simulated weight-sync
  collective.
"""
from __future__ import annotations

from typing import Dict, List, Optional, Sequence, Tuple

import numpy as np

from ..graphs import BWD, FWD, CompGraph, partitioned_name


class _MutableGraph:
    """Insertion-ordered digraph used only inside the (memoised) partition
    transform; mirrors networkx MultiDiGraph iteration-order semantics."""

    def __init__(self):
        self.nodes: "Dict[str, dict]" = {}
        self.succ: "Dict[str, Dict[str, None]]" = {}
        self.pred: "Dict[str, Dict[str, None]]" = {}
        self.edge_attr: "Dict[Tuple[str, str], dict]" = {}

    def add_node(self, name: str, **attrs):
        if name not in self.nodes:
            self.nodes[name] = {}
            self.succ[name] = {}
            self.pred[name] = {}
        self.nodes[name].update(attrs)

    def add_edge(self, u: str, v: str, **attrs):
        self.succ[u][v] = None
        self.pred[v][u] = None
        self.edge_attr.setdefault((u, v), {}).update(attrs)

    def remove_node(self, name: str):
        for v in list(self.succ[name]):
            del self.pred[v][name]
            self.edge_attr.pop((name, v), None)
        for u in list(self.pred[name]):
            del self.succ[u][name]
            self.edge_attr.pop((u, name), None)
        del self.succ[name]
        del self.pred[name]
        del self.nodes[name]

    def edges(self) -> List[Tuple[str, str]]:
        return [(u, v) for u in self.nodes for v in self.succ[u]]

    def to_comp_graph(self, model: str) -> CompGraph:
        names = list(self.nodes.keys())
        idx = {nm: i for i, nm in enumerate(names)}
        device_types = list(next(iter(self.nodes.values()))["compute_cost"].keys())
        comp = {dt: np.array([self.nodes[nm]["compute_cost"][dt] for nm in names])
                for dt in device_types}
        mem = np.array([self.nodes[nm]["memory_cost"] for nm in names])
        ptype = np.array([FWD if self.nodes[nm]["pass_type"] == "forward_pass" else BWD
                          for nm in names])
        edges = self.edges()
        src = np.array([idx[u] for u, _ in edges], dtype=np.int64)
        dst = np.array([idx[v] for _, v in edges], dtype=np.int64)
        size = np.array([self.edge_attr[e].get("size", 0.0) for e in edges])
        counterpart = np.full(len(names), -1, dtype=np.int64)
        return CompGraph(names=names, compute_cost=comp, memory_cost=mem,
                         pass_type=ptype, counterpart=counterpart,
                         src=src, dst=dst, size=size, model=model)


def _mutable_from_comp_graph(g: CompGraph) -> _MutableGraph:
    mg = _MutableGraph()
    for i, nm in enumerate(g.names):
        mg.add_node(nm,
                    compute_cost={dt: float(cc[i]) for dt, cc in g.compute_cost.items()},
                    memory_cost=float(g.memory_cost[i]),
                    pass_type="forward_pass" if g.pass_type[i] == FWD else "backward_pass")
    for e in range(g.m):
        mg.add_edge(g.names[int(g.src[e])], g.names[int(g.dst[e])],
                    size=float(g.size[e]))
    return mg


def data_split(g: CompGraph, dp_splits: int = 0) -> _MutableGraph:
    """Reference ``data_split_node:5-40``: dp_splits+1 shifted clones; every edge
    size becomes the source node's memory_cost."""
    og_idx = list(range(g.n))
    highest = max(int(nm) for nm in g.names)
    mg = _MutableGraph()
    for i in range(dp_splits + 1):
        shift = i * highest
        for j in og_idx:
            mg.add_node(str(int(g.names[j]) + shift),
                        compute_cost={dt: float(cc[j]) for dt, cc in g.compute_cost.items()},
                        memory_cost=float(g.memory_cost[j]),
                        pass_type="forward_pass" if g.pass_type[j] == FWD else "backward_pass")
        for e in range(g.m):
            u = str(int(g.names[int(g.src[e])]) + shift)
            v = str(int(g.names[int(g.dst[e])]) + shift)
            mg.add_edge(u, v)
    for (u, v) in mg.edges():
        mg.edge_attr[(u, v)]["size"] = mg.nodes[u]["memory_cost"]
    return mg


def model_split(mg: _MutableGraph,
                mp_split_ids: Sequence,
                mp_splits: Sequence[int],
                num_fwd_nodes: int,
                dp_splits: int = 0) -> _MutableGraph:
    """Reference ``model_split_node:42-110`` on a mutable graph in place."""
    highest = num_fwd_nodes * 2  # highest original node id after mirroring
    in_edge_features: Dict[Tuple[str, str], dict] = {}
    out_edge_features: Dict[Tuple[str, str], dict] = {}

    for i in range(len(mp_split_ids)):
        sid = str(mp_split_ids[i])
        if sid not in mg.nodes:
            continue
        if mg.nodes[sid]["pass_type"] != "forward_pass":
            continue  # backward ops handled together with their forward twin
        n_splits = int(mp_splits[i])
        for j in range(dp_splits + 1):
            node_ids = [
                str(int(sid) + j * highest),
                str(highest - (int(sid) - 1) + j * highest),  # backward twin
            ]
            for k, node_id in enumerate(node_ids):
                in_nbrs = list(mg.pred[node_id].keys())
                out_nbrs = list(mg.succ[node_id].keys())
                old = mg.nodes[node_id]
                new_feature = {
                    "compute_cost": {dt: c / n_splits for dt, c in old["compute_cost"].items()},
                    "memory_cost": old["memory_cost"] / n_splits,
                    "pass_type": old["pass_type"],
                }
                new_names = [partitioned_name(node_id, s) for s in range(n_splits)]
                new_edges: List[Tuple[str, str]] = []
                for nn in new_names:
                    for p in in_nbrs:
                        new_edges.append((p, nn))
                        in_edge_features[(p, nn)] = {
                            "size": mg.nodes[p]["memory_cost"] / n_splits}
                    for c in out_nbrs:
                        new_edges.append((nn, c))
                        out_edge_features[(nn, c)] = {
                            "size": mg.nodes[c]["memory_cost"] / n_splits}
                if k == 1:  # backward: all-to-all weight-sync edges
                    for l in range(n_splits):
                        for m_ in range(n_splits):
                            if l == m_:
                                continue
                            new_edges.append((new_names[l], new_names[m_]))
                            in_edge_features[(new_names[l], new_names[m_])] = {
                                "size": new_feature["memory_cost"]}
                mg.remove_node(node_id)
                for nn in new_names:
                    mg.add_node(nn, **{**new_feature,
                                       "compute_cost": dict(new_feature["compute_cost"])})
                for (u, v) in new_edges:
                    mg.add_edge(u, v)

    # apply in-features first, then out-features (out wins on conflicts);
    # stale entries for removed edges are ignored (networkx semantics)
    for feats in (in_edge_features, out_edge_features):
        for (u, v), attr in feats.items():
            if (u, v) in mg.edge_attr:
                mg.edge_attr[(u, v)].update(attr)
    return mg


def build_partitioned_graph(g: CompGraph,
                            mp_split_ids: Sequence,
                            mp_splits: Sequence[int],
                            dp_splits: int = 0,
                            model: Optional[str] = None) -> CompGraph:
    """data_split (dp) then model_split (mp), as ``OpPartition`` does
    (``actions/op_partition.py:52-64``)."""
    num_fwd = int((g.pass_type == FWD).sum())
    mg = data_split(g, dp_splits=dp_splits)
    mg = model_split(mg, mp_split_ids, mp_splits, num_fwd_nodes=num_fwd,
                     dp_splits=dp_splits)
    return mg.to_comp_graph(model=model if model is not None else g.model)
