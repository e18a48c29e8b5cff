"""Native RCCL layer: ordered dispatcher + order agreement (CPU tests).

The GPU-side launches need hardware (tests/test_gpu_rccl.py); the ordering
and agreement machinery (csrc/rccl/dispatcher.hpp + scheduler_agree over
the control plane) is hardware-independent and verified here across real
processes, including adversarial per-rank arrival orders
(reference: srcs/cpp/src/nccl/scheduler.cpp).
"""
import pytest

from tests.mp_helpers import (rccl_local_scope_agreement_body,
                              rccl_order_agreement_body, retry_flaky,
                              spawn_cluster)


def _has_rccl():
    try:
        import kungfu_amd._rccl  # noqa: F401

        return True
    except ImportError:
        return False


pytestmark = pytest.mark.skipif(not _has_rccl(),
                                reason="_rccl extension not built")


@retry_flaky
def _run_order(port_block):
    res = spawn_cluster(rccl_order_agreement_body, 3, port_block)
    assert res == [True] * 3


@retry_flaky
def _run_local(port_block):
    res = spawn_cluster(rccl_local_scope_agreement_body, 2, port_block)
    assert res == [True] * 2


def test_rccl_order_agreement(port_block):
    _run_order(port_block)


def test_rccl_local_scope_agreement(port_block):
    _run_local(port_block)
