"""Property-based invariants (hypothesis) for the graph IR and the
stage partitioner — randomized breadth on top of the fixed-case tests
(the reference's graph/test.py has only committed fixtures)."""

import pytest
from hypothesis import given, settings, strategies as st

from ddlbench_amd.parallel.pipeline.graph import (Graph, Node,
                                                  compress_branches)
from ddlbench_amd.parallel.pipeline.partition import (partition_chain,
                                                      partition_graph)

settings.register_profile("ci", max_examples=25, deadline=None)
settings.load_profile("ci")


def _chain_graph(times):
    return Graph.chain([
        Node(i, f"L{i}", fwd_time=t / 3, bwd_time=2 * t / 3,
             activation_size=64.0, parameter_size=32.0)
        for i, t in enumerate(times)])


@st.composite
def random_dag(draw):
    """Small random DAG: edges only forward in id order, connected
    enough to have a single source chain skeleton."""
    n = draw(st.integers(min_value=2, max_value=7))
    g = Graph()
    for i in range(n):
        t = draw(st.floats(min_value=0.01, max_value=4.0))
        g.add_node(Node(i, f"L{i}", fwd_time=t / 2, bwd_time=t / 2,
                        activation_size=16.0, parameter_size=8.0))
    for i in range(1, n):
        # at least one parent among earlier nodes keeps it connected
        parents = draw(st.sets(st.integers(0, i - 1), min_size=1,
                               max_size=min(i, 3)))
        for p in parents:
            g.add_edge(p, i)
    return g


@given(times=st.lists(st.floats(min_value=0.01, max_value=10.0),
                      min_size=1, max_size=10),
       m=st.integers(min_value=1, max_value=4))
def test_partition_chain_invariants(times, m):
    g = _chain_graph(times)
    res = partition_chain(g, m, bw=1e12, memory_bytes=1e15)
    # full contiguous cover, in order
    covered = [l for s in res.stages for l in s.layers]
    assert covered == list(range(len(times)))
    # replicas sum to exactly the GPU count
    assert sum(s.replicas for s in res.stages) == m
    # bottleneck can never beat the ideal work/M lower bound
    total = sum(times)
    assert res.bottleneck >= total / m - 1e-9
    # one stage on one GPU == the whole chain time
    if m == 1:
        assert len(res.stages) == 1
        assert res.bottleneck == pytest.approx(total, rel=1e-6)


@given(g=random_dag(), m=st.integers(min_value=1, max_value=3))
def test_partition_dag_invariants(g, m):
    n = len(g.nodes)
    total = sum(nd.compute_time for nd in g.nodes.values())
    res = partition_graph(g, m, bw=1e12, memory_bytes=1e15)
    covered = sorted(l for s in res.stages for l in s.layers)
    assert covered == list(range(n))
    assert sum(s.replicas for s in res.stages) == m
    assert res.bottleneck >= total / m - 1e-9
    # convexity: stage ids never decrease along any edge
    for src, dsts in g.edges.items():
        for dst in dsts:
            assert g.nodes[src].stage_id <= g.nodes[dst].stage_id


@given(g=random_dag())
def test_compress_branches_preserves_work(g):
    total_t = sum(nd.compute_time for nd in g.nodes.values())
    total_p = sum(nd.parameter_size for nd in g.nodes.values())
    cg, mapping = compress_branches(g)
    # mapping partitions the original node set exactly
    flat = sorted(i for v in mapping.values() for i in v)
    assert flat == sorted(g.nodes.keys())
    # compute and parameter totals survive compression
    assert sum(nd.compute_time for nd in cg.nodes.values()) == \
        pytest.approx(total_t, rel=1e-9)
    assert sum(nd.parameter_size for nd in cg.nodes.values()) == \
        pytest.approx(total_p, rel=1e-9)
    # the compressed graph is still a DAG
    cg.topological_sort()


@given(g=random_dag())
def test_graph_serialization_roundtrip(g):
    g2 = Graph.loads(g.dumps())
    assert sorted(g2.nodes.keys()) == sorted(g.nodes.keys())
    for i, nd in g.nodes.items():
        assert g2.nodes[i].fwd_time == pytest.approx(nd.fwd_time)
        assert g2.nodes[i].desc == nd.desc
    assert {(s, d) for s, ds in g.edges.items() for d in ds} == \
        {(s, d) for s, ds in g2.edges.items() for d in ds}


@given(words=st.lists(
    st.text(alphabet=st.characters(whitelist_categories=("Ll", "Lu"),
                                   max_codepoint=0x24F),
            min_size=1, max_size=6),
    min_size=1, max_size=8))
def test_tokenizer_roundtrip_property(words):
    from ddlbench_amd.data.tokenizer import Tokenizer
    line = " ".join(words)
    tok = Tokenizer.build([line])
    ids = tok.encode(line)
    assert tok.decode(ids) == " ".join(line.split())
