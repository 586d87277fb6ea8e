"""Graph reader / mirroring / flat-structure tests.

Semantics under test mirror reference ``ddls/utils.py:269-480``
(pipedream reader + mirror_graph + combine_graphs).
"""
import numpy as np
import pytest

from ddls_amd.graphs import (FWD, BWD, backward_name, base_name_and_split,
                             load_pipedream_graph, partitioned_name)


@pytest.fixture
def tiny_graph(tiny_model_files):
    return load_pipedream_graph(tiny_model_files + "/tiny.txt", "A100")


def test_mirrored_structure(tiny_graph):
    g = tiny_graph
    # 2 fwd + 2 bwd nodes; edges: fwd 1->2, join 2->3, bwd 3->4
    assert g.n == 4
    assert g.m == 3
    assert g.names == ["1", "2", "4", "3"]
    assert list(g.pass_type) == [FWD, FWD, BWD, BWD]
    edges = set(zip((g.names[int(u)] for u in g.src),
                    (g.names[int(v)] for v in g.dst)))
    assert edges == {("1", "2"), ("2", "3"), ("3", "4")}


def test_costs_and_sizes(tiny_graph):
    g = tiny_graph
    i1, i2 = g.name_to_idx["1"], g.name_to_idx["2"]
    b1, b2 = g.name_to_idx["4"], g.name_to_idx["3"]
    cc = g.compute_cost["A100"]
    assert cc[i1] == pytest.approx(0.02)
    assert cc[i2] == pytest.approx(0.03)
    assert cc[b1] == pytest.approx(0.04)  # backward of op 1
    assert cc[b2] == pytest.approx(0.06)
    # memory = activation + parameter
    assert g.memory_cost[i1] == pytest.approx(1.5e8)
    assert g.memory_cost[b2] == pytest.approx(3e8)
    # edge size = activation of source's forward op
    size = {(g.names[int(u)], g.names[int(v)]): s
            for u, v, s in zip(g.src, g.dst, g.size)}
    assert size[("1", "2")] == pytest.approx(1e8)
    assert size[("2", "3")] == pytest.approx(2e8)   # join: activation of op 2
    assert size[("3", "4")] == pytest.approx(2e8)   # bwd edge: activation of op 2


def test_counterpart_and_names(tiny_graph):
    g = tiny_graph
    assert g.names[int(g.counterpart[g.name_to_idx["1"]])] == "4"
    assert g.names[int(g.counterpart[g.name_to_idx["2"]])] == "3"
    assert backward_name("1", 2) == "4"
    assert backward_name("2", 2) == "3"
    assert partitioned_name("3", 0) == "3a"
    assert partitioned_name("3", 2) == "3c"
    assert base_name_and_split("3c") == ("3", 2)
    assert base_name_and_split("17") == ("17", None)


def test_depths_and_topo(tiny_graph):
    g = tiny_graph
    d = g.node_depths()
    assert d[g.name_to_idx["1"]] == 1
    assert d[g.name_to_idx["2"]] == 2
    assert d[g.name_to_idx["3"]] == 3
    assert d[g.name_to_idx["4"]] == 4
    order = [g.names[i] for i in g.topo_order_fwd_subgraph()]
    assert order == ["1", "2"]


def test_true_parent_count_excludes_bidirectional():
    from ddls_amd.graphs import CompGraph
    # a <-> b sync pair plus c -> b
    g = CompGraph(names=["a", "b", "c"],
                  compute_cost={"X": np.array([1.0, 1.0, 1.0])},
                  memory_cost=np.array([1.0, 1.0, 1.0]),
                  pass_type=np.array([BWD, BWD, BWD]),
                  counterpart=np.array([-1, -1, -1]),
                  src=np.array([0, 1, 2]), dst=np.array([1, 0, 1]),
                  size=np.array([1.0, 1.0, 1.0]))
    tpc = g.true_parent_count
    assert tpc[g.name_to_idx["b"]] == 1  # only c; a<->b bidirectional excluded
    assert tpc[g.name_to_idx["a"]] == 0
