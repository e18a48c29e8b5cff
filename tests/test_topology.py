"""Topology generator unit tests (reference: srcs/go/plan/topology_test.go,
graph/graph_test.go)."""
import pytest

from kungfu_amd import _core

STRATEGIES = [
    "STAR", "MULTI_STAR", "RING", "CLIQUE", "TREE", "BINARY_TREE",
    "BINARY_TREE_STAR", "MULTI_BINARY_TREE_STAR", "AUTO",
]


def check_reduce_bcast_pair(n, g):
    """Every strategy's reduce graph must funnel all ranks into exactly one
    root (with a self loop), and the bcast graph must reach every rank from
    that root."""
    redges = g["reduce"]
    bedges = g["bcast"]
    roots = g["roots"]
    assert len(roots) >= 1
    # reduce: out-degree <= 1 per rank except root; all paths end at a root
    nexts = {}
    for a, b in redges:
        nexts.setdefault(a, []).append(b)
    for r in range(n):
        v, hops = r, 0
        while v not in roots and hops <= n:
            outs = nexts.get(v, [])
            assert len(outs) == 1, "rank %d has %d parents" % (v, len(outs))
            v = outs[0]
            hops += 1
        assert v in roots, "rank %d does not reach a root" % r
    # bcast: BFS from roots reaches everyone
    adj = {}
    for a, b in bedges:
        adj.setdefault(a, []).append(b)
    seen = set(roots)
    frontier = list(roots)
    while frontier:
        v = frontier.pop()
        for u in adj.get(v, []):
            if u not in seen:
                seen.add(u)
                frontier.append(u)
    assert seen == set(range(n)), "bcast does not reach all ranks"


@pytest.mark.parametrize("strategy", STRATEGIES)
@pytest.mark.parametrize("n", [1, 2, 3, 4, 7, 8])
def test_single_host_strategies(strategy, n):
    gs = _core.topology_edges(n, strategy, "")
    assert len(gs) >= 1
    for g in gs:
        check_reduce_bcast_pair(n, g)


@pytest.mark.parametrize("strategy", STRATEGIES)
def test_multi_host_strategies(strategy):
    # 2 hosts x 4 slots
    peers = ",".join(["10.0.0.1:%d" % (31100 + i) for i in range(4)] +
                     ["10.0.0.2:%d" % (31100 + i) for i in range(4)])
    gs = _core.topology_edges(0, strategy, peers)
    for g in gs:
        check_reduce_bcast_pair(8, g)


def test_ring_has_n_rotations():
    gs = _core.topology_edges(5, "RING", "")
    assert len(gs) == 5
    roots = [g["roots"][0] for g in gs]
    assert sorted(roots) == list(range(5))  # each rotation has its own root


def test_clique_has_n_stars():
    gs = _core.topology_edges(6, "CLIQUE", "")
    assert len(gs) == 6
    assert sorted(g["roots"][0] for g in gs) == list(range(6))


def test_auto_is_star_on_one_host():
    a = _core.topology_digest("127.0.0.1:1,127.0.0.1:2", "AUTO")
    s = _core.topology_digest("127.0.0.1:1,127.0.0.1:2", "STAR")
    assert a == s


def test_digest_stability():
    peers = "127.0.0.1:1,127.0.0.1:2,127.0.0.1:3,127.0.0.1:4"
    assert (_core.topology_digest(peers, "BINARY_TREE") ==
            _core.topology_digest(peers, "BINARY_TREE"))
    assert (_core.topology_digest(peers, "BINARY_TREE") !=
            _core.topology_digest(peers, "STAR"))
