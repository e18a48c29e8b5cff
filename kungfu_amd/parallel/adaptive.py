"""Adaptive communication: interference voting + collective strategy
switching + latency-MST topology install.

Reference parity: srcs/go/kungfu/session/adaptiveStrategies.go — every
monitored window each peer votes (throughput < ratio * best) via an i32
sum all-reduce; a majority vote makes ALL peers switch the global strategy
(the vote result is identical everywhere, so the switch is itself a
consensus decision), mirroring CheckInterference + SetGlobalStrategy
(session/adaptation.go:8-28).
"""
import numpy as np

from kungfu_amd import _core, _ensure_init

# escalation order when interference is detected (reference AP.Strategies)
_FALLBACK_ORDER = [
    "BINARY_TREE_STAR", "MULTI_BINARY_TREE_STAR", "TREE", "BINARY_TREE",
    "STAR", "MULTI_STAR", "RING", "CLIQUE",
]


def vote_interference(ratio=0.8, name="|interference"):
    """Return (my_vote, total_votes) across the cluster."""
    _ensure_init()
    mine = 1 if _core.check_interference(ratio) else 0
    buf = np.array([mine], dtype=np.int32)
    out = np.zeros(1, dtype=np.int32)
    _core.all_reduce(buf.ctypes.data, out.ctypes.data, 1, 3, 0, name)
    return mine, int(out[0])


def check_interference_and_switch(ratio=0.8):
    """One adaptation round: majority interference vote -> rotate to the
    next strategy in the fallback order on every peer. Returns the new
    strategy name, or None if no switch happened."""
    _ensure_init()
    if _core.size() <= 1:
        return None
    _, votes = vote_interference(ratio)
    if votes * 2 <= _core.size():
        return None
    cur = _core.get_strategy()
    try:
        idx = _FALLBACK_ORDER.index(cur)
    except ValueError:
        idx = -1
    nxt = _FALLBACK_ORDER[(idx + 1) % len(_FALLBACK_ORDER)]
    # every peer computed the same `votes`, hence the same `nxt`
    _core.set_strategy(nxt)
    _core.reset_strategy_stats()
    return nxt


def install_mst_topology():
    """Measure pairwise latencies and install the minimum spanning tree as
    the collective topology (reference ops/adapt.py:49-60)."""
    from kungfu_amd.ops import compute_mst_tree

    return compute_mst_tree()


def strategy_throughputs():
    """Per-strategy monitored throughput stats (bytes/sec)."""
    _ensure_init()
    return _core.strategy_stats()


def print_strategy_stats():
    """Log per-strategy monitored throughput (reference PrintStategyStats,
    session/adaptiveStrategies.go)."""
    from kungfu_amd import rank

    stats = strategy_throughputs()
    for i, s in enumerate(stats):
        print("[kungfu rank %d] strategy %d: ops=%d bytes=%d "
              "throughput=%.1f MB/s" %
              (rank(), i, s["ops"], s["bytes"], s["throughput"] / 1e6),
              flush=True)
    return stats
