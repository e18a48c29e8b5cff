"""Native RCCL layer on hardware (gfx950).

Single-device coverage: the full bootstrap (uniqueId over the control
plane), every collective, stream-ordered completion, reinit. True
multi-device runs happen in the driver's multi-GPU bench; the two-ranks-
one-device probe documents RCCL's duplicate-device behavior.
"""
import pytest
import torch

from mp_helpers import (rccl_gpu_pair_body, rccl_gpu_world1_body,
                        spawn_cluster)

pytestmark = pytest.mark.gpu


@pytest.mark.skipif(not torch.cuda.is_available(), reason="needs GPU")
def test_rccl_native_world1(port_block):
    res = spawn_cluster(rccl_gpu_world1_body, 1, port_block, timeout=300)
    assert res == [True]


@pytest.mark.skipif(not torch.cuda.is_available(), reason="needs GPU")
@pytest.mark.skipif(torch.cuda.is_available() and
                    torch.cuda.device_count() < 1, reason="needs GPU")
def test_rccl_native_two_ranks_one_device(port_block):
    res = spawn_cluster(rccl_gpu_pair_body, 2, port_block,
                        extra={"HIP_VISIBLE_DEVICES": "0",
                               "CUDA_VISIBLE_DEVICES": "0"},
                        timeout=300)
    assert all(r == "ok" or str(r).startswith("unsupported") for r in res), res
    print("two-ranks-one-device:", res)


@pytest.mark.skipif(not torch.cuda.is_available(), reason="needs GPU")
def test_rccl_async_storm(port_block):
    from mp_helpers import rccl_gpu_storm_body

    res = spawn_cluster(rccl_gpu_storm_body, 1, port_block, timeout=300)
    assert res == [True]


@pytest.mark.skipif(not torch.cuda.is_available(), reason="needs GPU")
def test_rccl_graph_capture_inline(port_block):
    from mp_helpers import rccl_gpu_graph_capture_body

    res = spawn_cluster(rccl_gpu_graph_capture_body, 1, port_block,
                        timeout=300)
    assert res == [True]
