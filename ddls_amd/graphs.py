"""Flat-array computation graphs.

The reference (cwfparsonson/ddls) represents a job's computation graph as an
``nx.MultiDiGraph`` with per-node/per-edge attribute dicts
(``ddls/demands/jobs/job.py:42``, ``ddls/utils.py:269-480``).  This rebuild uses a
struct-of-arrays layout instead: node/edge attributes are contiguous numpy arrays
indexed by dense integer ids.  That makes the simulator's tick loops vectorisable on
CPU and directly uploadable to GPU HBM for batched HIP env stepping.

External (reference-compatible) op ids are strings ("1".."2n" for the mirrored
graph, "3a","3b",... for partitioned sub-ops); they live in ``names`` and are only
used at API boundaries.
"""
from __future__ import annotations

import json
import re
from dataclasses import dataclass
from typing import Dict, List, Optional, Sequence, Tuple

import numpy as np

FWD = 0
BWD = 1


class CompGraph:
    """Immutable computation-graph structure (shared between jobs of one model).

    Mirrors the attribute vocabulary of the reference's ddls graphs
    (``ddls/utils.py:402-480``): per-node ``compute_cost`` (dict device_type ->
    cost), ``memory_cost``, ``pass_type``, ``forward_node_id``/``backward_node_id``;
    per-edge ``size``.
    """

    __slots__ = (
        "names", "name_to_idx", "compute_cost", "memory_cost", "pass_type",
        "counterpart", "src", "dst", "size", "n", "m",
        "_out_csr", "_in_csr", "_true_parent_count", "_bidir_edge",
        "_reverse_edge", "model",
    )

    def __init__(self,
                 names: Sequence[str],
                 compute_cost: Dict[str, np.ndarray],
                 memory_cost: np.ndarray,
                 pass_type: np.ndarray,
                 counterpart: np.ndarray,
                 src: np.ndarray,
                 dst: np.ndarray,
                 size: np.ndarray,
                 model: str = ""):
        self.names = list(names)
        self.name_to_idx = {nm: i for i, nm in enumerate(self.names)}
        self.compute_cost = {k: np.asarray(v, dtype=np.float64) for k, v in compute_cost.items()}
        self.memory_cost = np.asarray(memory_cost, dtype=np.float64)
        self.pass_type = np.asarray(pass_type, dtype=np.int8)
        # counterpart[i] = index of the mirrored op (fwd<->bwd), or -1
        self.counterpart = np.asarray(counterpart, dtype=np.int64)
        self.src = np.asarray(src, dtype=np.int64)
        self.dst = np.asarray(dst, dtype=np.int64)
        self.size = np.asarray(size, dtype=np.float64)
        self.n = len(self.names)
        self.m = len(self.src)
        self.model = model
        self._out_csr = None
        self._in_csr = None
        self._true_parent_count = None
        self._bidir_edge = None
        self._reverse_edge = None

    def __deepcopy__(self, memo):
        # immutable: share across deep copies (jobs of one model share structure)
        return self

    # ---------------- adjacency ----------------
    def _build_csr(self, key_arr):
        order = np.argsort(key_arr, kind="stable")
        counts = np.bincount(key_arr, minlength=self.n)
        indptr = np.zeros(self.n + 1, dtype=np.int64)
        np.cumsum(counts, out=indptr[1:])
        return indptr, order

    @property
    def out_csr(self):
        if self._out_csr is None:
            self._out_csr = self._build_csr(self.src)
        return self._out_csr

    @property
    def in_csr(self):
        if self._in_csr is None:
            self._in_csr = self._build_csr(self.dst)
        return self._in_csr

    def out_edges_of(self, node: int) -> np.ndarray:
        indptr, order = self.out_csr
        return order[indptr[node]:indptr[node + 1]]

    def in_edges_of(self, node: int) -> np.ndarray:
        indptr, order = self.in_csr
        return order[indptr[node]:indptr[node + 1]]

    @property
    def bidir_edge(self) -> np.ndarray:
        """bool[m]: edge (u,v) for which (v,u) also exists (sync edges)."""
        if self._bidir_edge is None:
            pairs = set(zip(self.src.tolist(), self.dst.tolist()))
            self._bidir_edge = np.fromiter(
                ((v, u) in pairs for u, v in zip(self.src.tolist(), self.dst.tolist())),
                dtype=bool, count=self.m)
        return self._bidir_edge

    @property
    def reverse_edge(self) -> np.ndarray:
        """int64[m]: index of edge (v,u) for edge e=(u,v), or -1."""
        if self._reverse_edge is None:
            lookup = {(int(u), int(v)): e
                      for e, (u, v) in enumerate(zip(self.src, self.dst))}
            self._reverse_edge = np.array(
                [lookup.get((int(v), int(u)), -1)
                 for u, v in zip(self.src, self.dst)], dtype=np.int64)
        return self._reverse_edge

    @property
    def true_parent_count(self) -> np.ndarray:
        """Number of non-bidirectional parents per node (reference
        ``job.py:508-523`` get_op_parents semantics)."""
        if self._true_parent_count is None:
            not_bidir = ~self.bidir_edge
            self._true_parent_count = np.bincount(
                self.dst[not_bidir], minlength=self.n).astype(np.int64)
        return self._true_parent_count

    # ---------------- derived quantities ----------------
    def source_nodes(self) -> np.ndarray:
        in_deg = np.bincount(self.dst, minlength=self.n)
        return np.flatnonzero(in_deg == 0)

    def node_depths(self) -> np.ndarray:
        """BFS path length (number of nodes incl. endpoints) from the first
        source node; 0 if unreachable (reference ``job.py:23-29,250-316``)."""
        srcs = self.source_nodes()
        depth = np.zeros(self.n, dtype=np.int64)
        if len(srcs) == 0:
            return depth
        root = int(srcs[0])
        depth[root] = 1
        frontier = [root]
        indptr, order = self.out_csr
        while frontier:
            nxt = []
            for u in frontier:
                for e in order[indptr[u]:indptr[u + 1]]:
                    v = int(self.dst[e])
                    if depth[v] == 0 and v != root:
                        depth[v] = depth[u] + 1
                        nxt.append(v)
            frontier = nxt
        return depth

    def topo_order_fwd_subgraph(self) -> List[int]:
        """Kahn topological order of the forward-pass subgraph, FIFO queue,
        seeded in node-id order (reference ``placers/utils.py:100-114``)."""
        fwd_nodes = [i for i in range(self.n) if self.pass_type[i] == FWD]
        fwd_set = set(fwd_nodes)
        parents_cnt = {i: 0 for i in fwd_nodes}
        children = {i: [] for i in fwd_nodes}
        for e in range(self.m):
            u, v = int(self.src[e]), int(self.dst[e])
            if u in fwd_set and v in fwd_set:
                parents_cnt[v] += 1
                children[u].append(v)
        from collections import deque
        seq, queue = [], deque()
        for nd in fwd_nodes:
            if parents_cnt[nd] == 0:
                queue.append(nd)
                seq.append(nd)
        while queue:
            nd = queue.popleft()
            for ch in children[nd]:
                parents_cnt[ch] -= 1
                if parents_cnt[ch] == 0:
                    queue.append(ch)
                    seq.append(ch)
        return seq

    def total_memory_cost(self) -> float:
        return float(self.memory_cost.sum())

    def total_dep_size(self) -> float:
        return float(self.size.sum())

    def edge_id_tuples(self) -> List[Tuple[str, str, int]]:
        return [(self.names[int(u)], self.names[int(v)], 0)
                for u, v in zip(self.src, self.dst)]


# ---------------------------------------------------------------------------
# pipedream .txt reader + forward/backward mirroring
# (behavioural parity with ddls/utils.py:269-480)
# ---------------------------------------------------------------------------

def parse_pipedream_txt(path: str) -> Tuple[List[Tuple[str, dict]], List[Tuple[str, str]]]:
    nodes, edges = [], []
    with open(path) as f:
        for line in f:
            parts = line.split(" -- ")
            parts = [p.split("\t")[-1] for p in parts]
            if len(parts) > 2:
                node_id = str(int(parts[0][4:]))
                op_type = parts[1].split("(")[0]
                feats = {"type": op_type}
                comp_mem = parts[2].split(", ")
                for i, key in enumerate(["forward", "backward", "activation", "parameter"]):
                    raw = comp_mem[i].split("=")[1].replace("\n", "").replace(";", ",")
                    val = json.loads(raw)
                    if isinstance(val, list):
                        val = float(np.sum(val))
                    feats[key] = float(val)
                nodes.append((node_id, feats))
            else:
                u = str(int(parts[0][4:]))
                v = str(int(parts[1][4:]))
                edges.append((u, v))
    return nodes, edges


def mirrored_graph_from_pipedream(nodes, edges, processor_type_profiled: str,
                                  model: str = "") -> CompGraph:
    """Build the forward+backward mirrored ddls graph from pipedream nodes/edges.

    Reference: ``mirror_graph``/``combine_graphs``/``ddls_graph_from_pipedream_graph``
    (``ddls/utils.py:342-465``).  Forward node i in [1..n]; backward node
    2n-(i-1); join edge n -> n+1; every edge's size = activation of its source
    node; node memory_cost = activation + parameter.
    """
    n = len(nodes)
    id_map = {nid: k for k, (nid, _) in enumerate(nodes)}  # pipedream id -> 0..n-1

    names: List[str] = []
    comp, mem, ptype, counterpart = [], [], [], []
    activation = {}

    # forward nodes, in file order
    for nid, ft in nodes:
        names.append(nid)
        comp.append(ft["forward"])
        mem.append(ft["activation"] + ft["parameter"])
        ptype.append(FWD)
        activation[nid] = ft["activation"]
    # backward nodes
    for nid, ft in nodes:
        bwd_name = str(2 * n - (int(nid) - 1))
        names.append(bwd_name)
        comp.append(ft["backward"])
        mem.append(ft["activation"] + ft["parameter"])
        ptype.append(BWD)
    for k in range(n):
        counterpart.append(n + k)
    for k in range(n):
        counterpart.append(k)

    name_to_idx = {nm: i for i, nm in enumerate(names)}
    src, dst, size = [], [], []
    for u, v in edges:  # forward edges
        src.append(name_to_idx[u])
        dst.append(name_to_idx[v])
        size.append(activation[u])
    # backward edges: (2n-(v-1)) -> (2n-(u-1)); size = activation of src bwd node,
    # i.e. activation of forward op v (combine_graphs: size = nodes[edge0]['activation'])
    for u, v in edges:
        bu = str(2 * n - (int(v) - 1))
        bv = str(2 * n - (int(u) - 1))
        src.append(name_to_idx[bu])
        dst.append(name_to_idx[bv])
        size.append(activation[v])
    # join edge: max fwd node -> min bwd node
    join_u = str(max(int(nm) for nm in names[:n]))
    join_v = str(min(int(nm) for nm in names[n:]))
    src.append(name_to_idx[join_u])
    dst.append(name_to_idx[join_v])
    # join edge size = activation of join_u
    size.append(activation[join_u])

    return CompGraph(names=names,
                     compute_cost={processor_type_profiled: np.array(comp)},
                     memory_cost=np.array(mem),
                     pass_type=np.array(ptype),
                     counterpart=np.array(counterpart),
                     src=np.array(src), dst=np.array(dst), size=np.array(size),
                     model=model)


def load_pipedream_graph(path: str, processor_type_profiled: str,
                         model: Optional[str] = None) -> CompGraph:
    nodes, edges = parse_pipedream_txt(path)
    if model is None:
        base = path.rstrip("/").split("/")[-1]
        if base == "graph.txt":
            model = path.rstrip("/").split("/")[-2]
        else:
            model = base.replace(".txt", "")
    return mirrored_graph_from_pipedream(nodes, edges, processor_type_profiled, model=model)


# ---------------------------------------------------------------------------
# REGAL CostGraphDef .pbtxt reader (reference ddls/utils.py:110-268)
# ---------------------------------------------------------------------------

def parse_pbtxt_nodes(path: str) -> List[dict]:
    """Parse a REGAL-style CostGraphDef text proto into node dicts with
    id / input_info (preceding node ids) / output_info (sizes) /
    control_input / compute_cost."""
    nodes: List[dict] = []
    node = None
    with open(path) as f:
        for raw in f:
            line = raw.replace(" ", "").replace("\n", "")
            if line == "node{":
                if node is not None:
                    nodes.append(node)
                node = {"input_info": [], "output_info": [],
                        "control_input": [], "compute_cost": 0}
            elif line == "}" or line == "":
                continue
            elif line.startswith("preceding_node"):
                node["input_info"].append(int(line.split(":", 1)[1]))
            elif line.startswith("preceding_port") or line.startswith("alias_input_port"):
                continue
            elif line.startswith("input_info") or line.startswith("output_info"):
                continue
            elif line.startswith("control_input"):
                node["control_input"].append(int(line.split(":", 1)[1]))
            elif line.startswith("compute_cost"):
                node["compute_cost"] = int(line.split(":", 1)[1])
            elif line.startswith("size"):
                node["output_info"].append(int(line.split(":", 1)[1]))
            elif line.startswith("id"):
                node["id"] = int(line.split(":", 1)[1])
            elif line.startswith("name"):
                if "_SOURCE" in line:
                    node["id"] = 0
            else:
                raise ValueError(f"unrecognised pbtxt line {line!r}")
    if node is not None:
        nodes.append(node)
    return nodes


def load_pbtxt_graph(path: str, processor_type_profiled: str,
                     model: Optional[str] = None) -> CompGraph:
    """Build a flat CompGraph from a REGAL CostGraphDef .pbtxt.  Dependency
    sizes are randomly sampled from the parent's output_info sizes (the
    reference's documented hack for DeepMind's ambiguous port mapping,
    ``utils.py:171-183``).  Nodes carry no fwd/bwd mirroring (legacy generic
    cluster workloads)."""
    import random as _random
    nodes = parse_pbtxt_nodes(path)
    ids = [n.get("id", i) for i, n in enumerate(nodes)]
    id_to_row = {nid: i for i, nid in enumerate(ids)}
    names = [str(nid) for nid in ids]
    comp = np.array([float(n.get("compute_cost", 0)) for n in nodes])
    mem = np.zeros(len(nodes))
    src, dst, size = [], [], []
    for n, nid in zip(nodes, ids):
        for parent in n["input_info"]:
            out_sizes = nodes[id_to_row[parent]]["output_info"]
            s = float(_random.choice(out_sizes)) if out_sizes else 0.0
            src.append(id_to_row[parent])
            dst.append(id_to_row[nid])
            size.append(s)
        for parent in n["control_input"]:
            src.append(id_to_row[parent])
            dst.append(id_to_row[nid])
            size.append(0.0)
    if model is None:
        model = path.rstrip("/").split("/")[-1].replace(".pbtxt", "")
    return CompGraph(names=names,
                     compute_cost={processor_type_profiled: comp},
                     memory_cost=mem,
                     pass_type=np.zeros(len(nodes), dtype=np.int8),
                     counterpart=np.full(len(nodes), -1, dtype=np.int64),
                     src=np.array(src, dtype=np.int64),
                     dst=np.array(dst, dtype=np.int64),
                     size=np.array(size), model=model)


def backward_name(forward_name: str, num_fwd_nodes: int) -> str:
    """Reference ``placers/utils.py:316-322``."""
    return str((2 * num_fwd_nodes) - (int(forward_name) - 1))


def partitioned_name(name: str, split_id: int) -> str:
    """Reference ``placers/utils.py:324-330``: '<int(name)><a|b|c...>'."""
    return str(int(name)) + chr(97 + split_id)


_SPLIT_RE = re.compile(r"^(\d+)([a-z])$")


def base_name_and_split(name: str) -> Tuple[str, Optional[int]]:
    m = _SPLIT_RE.match(name)
    if m:
        return m.group(1), ord(m.group(2)) - 97
    return name, None
