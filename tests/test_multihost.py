"""Runtime multi-host simulation: ranks on 127.0.0.1 vs 127.0.0.2 (the
whole 127/8 block is loopback), so host grouping, cross-host topologies,
hierarchical local/cross collectives, and the non-colocated P2P inline
path all execute for real (reference pattern: cluster-in-docker tests,
.github/workflows/cluster.yaml)."""
import pytest

from mp_helpers import (hier_subgroup_body, multihost_body,
                        spawn_multihost)


@pytest.mark.parametrize("strategy", ["AUTO", "BINARY_TREE_STAR", "RING",
                                      "MULTI_BINARY_TREE_STAR",
                                      "MULTI_STAR", "TREE", "CLIQUE",
                                      "BINARY_TREE", "STAR"])
def test_two_hosts_collectives(strategy, port_block):
    results = spawn_multihost(multihost_body, 4, port_block, strategy,
                              hosts=2)
    for r in results:
        assert r["hosts"] == 2
        assert r["local_size"] == 2
        assert r["hier"] == pytest.approx(4.0)
        assert r["p2p"] is True


def test_hierarchical_subgroups(port_block):
    """Sub-group hierarchical all-reduce (local reduce -> cross masters ->
    local bcast) over torch.distributed groups, 2 'hosts' x 2 ranks.
    gloo here; the same code runs RCCL groups on GPU clusters."""
    import os

    os.environ["KUNGFU_TORCH_BACKEND"] = "gloo"
    try:
        # spawn_multihost workers read env set in their own process; pass
        # backend through the spawned env instead
        import multiprocessing  # noqa: F401

        results = spawn_multihost(_hier_with_gloo, 4, port_block, "AUTO",
                                  hosts=2)
    finally:
        os.environ.pop("KUNGFU_TORCH_BACKEND", None)
    expect = float(sum(range(1, 5)))
    for out, out2 in results:
        assert out == expect
        assert out2 == 4.0


def _hier_with_gloo(rank, np):
    import os

    os.environ["KUNGFU_TORCH_BACKEND"] = "gloo"
    return hier_subgroup_body(rank, np)


def test_three_hosts(port_block):
    results = spawn_multihost(multihost_body, 6, port_block, "AUTO",
                              hosts=3)
    for r in results:
        assert r["hosts"] == 3
        assert r["hier"] == pytest.approx(6.0)
