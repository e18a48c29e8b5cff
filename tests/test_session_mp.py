"""Multi-process collective engine tests over loopback (the reference's
core pattern: real runtime, N local processes, no GPUs —
scripts/tests/run-integration-tests.sh sweeps np x strategies)."""
import pytest

from mp_helpers import (allreduce_body, hierarchical_body, monitoring_body,
                        p2p_body, settree_body, spawn_cluster)


@pytest.mark.parametrize("strategy",
                         ["STAR", "RING", "BINARY_TREE", "CLIQUE",
                          "BINARY_TREE_STAR"])
@pytest.mark.parametrize("np", [2, 4])
def test_allreduce_strategies(np, strategy, port_block):
    results = spawn_cluster(allreduce_body, np, port_block, strategy)
    expect_sum = sum(range(1, np + 1))
    for r in results:
        assert r["small_sum"] == pytest.approx(expect_sum)
        assert r["min"] == 5
        assert r["max"] == (np - 1) * 10 + 5
        assert r["bcast"] == 0.0
        assert r["gathered"] == list(range(np))
        assert r["consensus_ok"] is True
        assert r["consensus_diff"] is False
        assert r["bf16_sum"] == pytest.approx(
            sum(1.0 + i for i in range(np)))
    assert results[0]["rooted"] == list(range(np))


def test_allreduce_np1(port_block):
    results = spawn_cluster(allreduce_body, 1, port_block, "AUTO")
    assert results[0]["small_sum"] == 1.0


def test_p2p_store(port_block):
    assert spawn_cluster(p2p_body, 3, port_block) == [True, True, True]


def test_hierarchical(port_block):
    results = spawn_cluster(hierarchical_body, 4, port_block)
    assert all(r == pytest.approx(4.0) for r in results)


def test_set_tree(port_block):
    results = spawn_cluster(settree_body, 4, port_block)
    assert all(r == pytest.approx(4.0) for r in results)


def test_monitoring(port_block):
    results = spawn_cluster(monitoring_body, 2, port_block)
    for r in results:
        assert r["ops"] == 3
        assert r["lat_len"] == 2
        assert r["egress_nonzero"]
        assert r["interference"] is False


def test_torch_op_wrappers(port_block):
    from mp_helpers import ops_wrappers_body

    results = spawn_cluster(ops_wrappers_body, 2, port_block)
    for r in results:
        assert r["avg"] == pytest.approx(1.5)
        assert r["bcast_root1"] == 1.0  # broadcast from root=1
        assert r["gather"] == [0.0, 1.0]
        assert r["hier"] == pytest.approx(2.0)
        assert r["p2p"] is True
    assert results[0]["reduced"] == pytest.approx(2.0)


def test_async_handles(port_block):
    from mp_helpers import async_ops_body

    assert spawn_cluster(async_ops_body, 3, port_block) == [True] * 3


def test_all_dtypes_reduce(port_block):
    from mp_helpers import dtype_sweep_body

    assert spawn_cluster(dtype_sweep_body, 2, port_block) == [True, True]


def test_allreduce_np8(port_block):
    """Full-node-shaped CPU cluster (8 workers, the driver's max)."""
    results = spawn_cluster(allreduce_body, 8, port_block, "RING",
                            timeout=180)
    expect = sum(range(1, 9))
    for r in results:
        assert r["small_sum"] == pytest.approx(expect)
        assert r["gathered"] == list(range(8))


def test_async_stress(port_block):
    """24 concurrent async all-reduces x 3 rounds with barriers
    interleaved: shakes the name-keyed rendezvous under concurrency."""
    from mp_helpers import async_stress_body

    assert spawn_cluster(async_stress_body, 3, port_block,
                         timeout=180) == [True] * 3


@pytest.mark.parametrize("strategy", ["STAR", "RING"])
def test_shm_collective_path(strategy, port_block):
    """Opt-in /dev/shm data path for colocated collective chunks
    (KUNGFU_SHM_COLLECTIVES=1): payloads >= 64 KiB ride tmpfs files, the
    socket carries only references."""
    results = spawn_cluster(allreduce_body, 4, port_block, strategy,
                            extra={"KUNGFU_SHM_COLLECTIVES": "1"})
    expect_sum = sum(range(1, 5))
    for r in results:
        assert r["small_sum"] == pytest.approx(expect_sum)
        assert r["gathered"] == list(range(4))


def test_monitored_all_reduce_with_tree(port_block):
    from mp_helpers import monitored_all_reduce_body, spawn_cluster

    res = spawn_cluster(monitored_all_reduce_body, 3, port_block)
    assert res == [True] * 3
