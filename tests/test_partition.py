"""Partition transform tests (reference ``agents/partitioners/utils.py``)."""
import numpy as np
import pytest

from ddls_amd.cluster.partition import build_partitioned_graph, data_split
from ddls_amd.graphs import load_pipedream_graph


@pytest.fixture
def tiny_graph(tiny_model_files):
    return load_pipedream_graph(tiny_model_files + "/tiny.txt", "A100")


def test_data_split_edge_sizes(tiny_graph):
    mg = data_split(tiny_graph, dp_splits=0)
    # every edge size overwritten with the SOURCE node's memory cost
    assert mg.edge_attr[("1", "2")]["size"] == pytest.approx(1.5e8)
    assert mg.edge_attr[("2", "3")]["size"] == pytest.approx(3e8)
    assert mg.edge_attr[("3", "4")]["size"] == pytest.approx(3e8)


def test_model_split_structure(tiny_graph):
    # split op 1 (and its backward twin 4) 2 ways
    pg = build_partitioned_graph(tiny_graph, mp_split_ids=["1"], mp_splits=[2])
    names = set(pg.names)
    assert names == {"1a", "1b", "2", "3", "4a", "4b"}
    edges = {(pg.names[int(u)], pg.names[int(v)]): s
             for u, v, s in zip(pg.src, pg.dst, pg.size)}
    # fwd: 1a->2, 1b->2; join 2->3; bwd: 3->4a, 3->4b; sync 4a<->4b
    assert set(edges) == {("1a", "2"), ("1b", "2"), ("2", "3"),
                          ("3", "4a"), ("3", "4b"),
                          ("4a", "4b"), ("4b", "4a")}
    i1a = pg.name_to_idx["1a"]
    assert pg.compute_cost["A100"][i1a] == pytest.approx(0.01)   # 0.02 / 2
    assert pg.memory_cost[i1a] == pytest.approx(0.75e8)          # 1.5e8 / 2
    # out-edge size from split node = DESTINATION memory / n (reference rule)
    assert edges[("1a", "2")] == pytest.approx(3e8 / 2)          # mem(2)=3e8
    # in-edge of split bwd = parent memory / n
    assert edges[("3", "4a")] == pytest.approx(3e8 / 2)          # mem(3)=3e8
    # sync edge size = split op's (divided) memory cost
    assert edges[("4a", "4b")] == pytest.approx(1.5e8 / 2)


def test_model_split_adjacent_both_split(tiny_graph):
    pg = build_partitioned_graph(tiny_graph, mp_split_ids=["1", "2"],
                                 mp_splits=[2, 2])
    names = set(pg.names)
    assert names == {"1a", "1b", "2a", "2b", "3a", "3b", "4a", "4b"}
    edges = {(pg.names[int(u)], pg.names[int(v)]): s
             for u, v, s in zip(pg.src, pg.dst, pg.size)}
    # cross edges 1{a,b} x 2{a,b}
    for a in "ab":
        for b in "ab":
            assert ("1" + a, "2" + b) in edges
            assert ("3" + a, "4" + b) in edges
    # op 1 split first, op 2 second: edge (1a,2b) created during 2's split
    # as an in-edge -> size = mem(1a)/2 = (1.5e8/2)/2
    assert edges[("1a", "2b")] == pytest.approx(1.5e8 / 4)
    # sync pairs for both bwd ops
    assert ("3a", "3b") in edges and ("3b", "3a") in edges
    assert ("4a", "4b") in edges and ("4b", "4a") in edges
    # total edges: 4 (fwd cross) + 4 (join 2i->3j) + 4 (bwd cross) + 4 sync
    assert pg.m == 16


def test_partition_degree_must_be_even():
    from ddls_amd.cluster.actions import OpPartition

    class FakeQueue:
        jobs = {}

    class FakeCluster:
        job_queue = FakeQueue()
        job_model_to_max_num_partitions_to_init_details = {}

    with pytest.raises(ValueError):
        OpPartition({1: {"1": 3}}, cluster=FakeCluster())
