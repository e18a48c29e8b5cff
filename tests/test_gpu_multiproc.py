"""Cross-process GPU paths on a single device: two workers share cuda:0
(HIP_VISIBLE_DEVICES pinned) and exchange models through the P2P shm
store — the full device->host->shm->host->device pair-averaging loop
without needing multiple GPUs."""
import pytest
import torch

from mp_helpers import gpu_pair_store_body, spawn_cluster

pytestmark = pytest.mark.gpu


@pytest.mark.skipif(not torch.cuda.is_available(), reason="needs GPU")
def test_pair_averaging_store_two_procs_one_gpu(port_block):
    res = spawn_cluster(gpu_pair_store_body, 2, port_block,
                        extra={"HIP_VISIBLE_DEVICES": "0",
                               "CUDA_VISIBLE_DEVICES": "0"},
                        timeout=240)
    assert res == [True, True]
