"""Distributed optimizer semantics on CPU clusters (np=2) — the reference's
test_optimizers.py pattern: train a couple of steps with every optimizer
and check synchronization invariants numerically."""
import pytest

from mp_helpers import spawn_cluster


def _make_model(seed):
    import torch

    torch.manual_seed(seed)
    from kungfu_amd.models import SLP

    return SLP(in_features=32, classes=4)


def sync_sgd_body(rank, np):
    import torch
    import kungfu_amd as kf
    from kungfu_amd.ops import broadcast_model
    from kungfu_amd.optimizers import SynchronousSGDOptimizer

    kf.init(with_torch=False)
    model = _make_model(seed=100 + rank)  # different init per rank
    broadcast_model(model)  # sync from rank 0
    opt = SynchronousSGDOptimizer(
        torch.optim.SGD(model.parameters(), lr=0.1))
    torch.manual_seed(7 + rank)  # different data per rank
    losses = []
    for _ in range(3):
        x = torch.randn(8, 32)
        y = torch.randint(0, 4, (8,))
        opt.zero_grad()
        loss = torch.nn.functional.cross_entropy(model(x), y)
        loss.backward()
        opt.step()
        losses.append(float(loss))
    # all ranks must hold identical weights after synced steps
    w = model.fc.weight.detach().flatten()
    digest = [round(float(v), 6) for v in w[:5]]
    kf.finalize()
    return digest


def sma_body(rank, np):
    import torch
    import kungfu_amd as kf
    from kungfu_amd.optimizers import SynchronousAveragingOptimizer

    kf.init(with_torch=False)
    model = _make_model(seed=200 + rank)
    opt = SynchronousAveragingOptimizer(
        torch.optim.SGD(model.parameters(), lr=0.0), alpha=1.0)
    # lr=0 and alpha=1: one step should make all replicas exactly the mean
    opt.zero_grad()
    x = torch.randn(4, 32)
    loss = model(x).sum()
    loss.backward()
    opt.step()
    w = model.fc.weight.detach().flatten()
    digest = [round(float(v), 6) for v in w[:5]]
    kf.finalize()
    return digest


def pair_body(rank, np):
    import torch
    import kungfu_amd as kf
    from kungfu_amd.optimizers import PairAveragingOptimizer

    kf.init(with_torch=False)
    model = _make_model(seed=300 + rank)
    opt = PairAveragingOptimizer(
        torch.optim.SGD(model.parameters(), lr=0.01))
    torch.manual_seed(17 + rank)
    for _ in range(3):
        x = torch.randn(4, 32)
        y = torch.randint(0, 4, (4,))
        opt.zero_grad()
        torch.nn.functional.cross_entropy(model(x), y).backward()
        opt.step()
    kf.barrier()
    kf.finalize()
    return True


def ada_body(rank, np):
    import torch
    import kungfu_amd as kf
    from kungfu_amd.optimizers import AdaptiveSGDOptimizer

    kf.init(with_torch=False)
    model = _make_model(seed=400 + rank)
    opt = AdaptiveSGDOptimizer(
        torch.optim.SGD(model.parameters(), lr=0.05), change_step=2)
    torch.manual_seed(27 + rank)
    for _ in range(4):
        x = torch.randn(4, 32)
        y = torch.randint(0, 4, (4,))
        opt.zero_grad()
        torch.nn.functional.cross_entropy(model(x), y).backward()
        opt.step()
    assert opt.synced  # switched to S-SGD
    w = model.fc.weight.detach().flatten()
    digest = [round(float(v), 6) for v in w[:5]]
    kf.finalize()
    return digest


def gns_body(rank, np):
    import torch
    import kungfu_amd as kf
    from kungfu_amd.optimizers import MonitorGradientNoiseScaleOptimizer

    kf.init(with_torch=False)
    model = _make_model(seed=500)  # same init
    opt = MonitorGradientNoiseScaleOptimizer(
        torch.optim.SGD(model.parameters(), lr=0.05),
        device_batch_size=8)
    torch.manual_seed(37 + rank)
    for _ in range(3):
        x = torch.randn(8, 32)
        y = torch.randint(0, 4, (8,))
        opt.zero_grad()
        torch.nn.functional.cross_entropy(model(x), y).backward()
        opt.step()
    gns = opt.noise_scale
    kf.finalize()
    import math

    return not math.isnan(gns)


def gvar_body(rank, np):
    import torch
    import kungfu_amd as kf
    from kungfu_amd.optimizers import MonitorGradientVarianceOptimizer

    kf.init(with_torch=False)
    model = _make_model(seed=600)
    opt = MonitorGradientVarianceOptimizer(
        torch.optim.SGD(model.parameters(), lr=0.05))
    torch.manual_seed(47 + rank)
    for _ in range(2):
        x = torch.randn(8, 32)
        y = torch.randint(0, 4, (8,))
        opt.zero_grad()
        torch.nn.functional.cross_entropy(model(x), y).backward()
        opt.step()
    var = opt.variance
    kf.finalize()
    return var >= -1e-4  # variance is non-negative up to fp noise


def test_sync_sgd_identical_weights(port_block):
    a, b = spawn_cluster(sync_sgd_body, 2, port_block)
    assert a == b


def test_sma_replicas_converge(port_block):
    a, b = spawn_cluster(sma_body, 2, port_block)
    assert a == pytest.approx(b, abs=1e-5)


def test_pair_averaging_runs(port_block):
    assert spawn_cluster(pair_body, 2, port_block) == [True, True]


def test_ada_sgd_switch(port_block):
    a, b = spawn_cluster(ada_body, 2, port_block)
    assert a == b  # after the switch + rebroadcast, replicas are identical


def test_gradient_noise_scale(port_block):
    assert all(spawn_cluster(gns_body, 2, port_block))


def test_gradient_variance(port_block):
    assert all(spawn_cluster(gvar_body, 2, port_block))


def sma_gns_body(rank, np):
    import math
    import torch
    import kungfu_amd as kf
    from kungfu_amd.optimizers import (GradNoiseScaleProbe,
                                       SynchronousAveragingOptimizer)

    kf.init(with_torch=False)
    model = _make_model(seed=700)
    opt = SynchronousAveragingOptimizer(
        torch.optim.SGD(model.parameters(), lr=0.05), alpha=0.5)
    probe = GradNoiseScaleProbe(model.parameters(), device_batch_size=8,
                                interval=1)
    torch.manual_seed(57 + rank)
    for _ in range(3):
        x = torch.randn(8, 32)
        y = torch.randint(0, 4, (8,))
        opt.zero_grad()
        torch.nn.functional.cross_entropy(model(x), y).backward()
        probe.observe()
        opt.step()
    kf.finalize()
    return not math.isnan(probe.noise_scale)


def test_sma_with_gns_probe(port_block):
    assert all(spawn_cluster(sma_gns_body, 2, port_block))


def grad_accum_body(rank, np):
    import torch
    import kungfu_amd as kf
    from kungfu_amd.ops import broadcast_model
    from kungfu_amd.optimizers import SynchronousSGDOptimizer

    kf.init(with_torch=False)
    model = _make_model(seed=900 + rank)
    broadcast_model(model)
    # overlap=False supports gradient accumulation (2 backwards per step)
    opt = SynchronousSGDOptimizer(
        torch.optim.SGD(model.parameters(), lr=0.1), overlap=False)
    torch.manual_seed(77 + rank)
    for _ in range(2):
        opt.zero_grad()
        for _ in range(2):  # micro-batches
            x = torch.randn(4, 32)
            y = torch.randint(0, 4, (4,))
            torch.nn.functional.cross_entropy(model(x), y).backward()
        opt.step()
    w = model.fc.weight.detach().flatten()
    digest = [round(float(v), 6) for v in w[:5]]
    kf.finalize()
    return digest


def test_grad_accumulation_no_overlap(port_block):
    a, b = spawn_cluster(grad_accum_body, 2, port_block)
    assert a == b  # replicas identical after accumulated synced steps


def gns_zero_noise_body(rank, np):
    import torch
    import kungfu_amd as kf
    from kungfu_amd.optimizers import MonitorGradientNoiseScaleOptimizer

    kf.init(with_torch=False)
    model = _make_model(seed=950)      # identical init
    opt = MonitorGradientNoiseScaleOptimizer(
        torch.optim.SGD(model.parameters(), lr=0.0), device_batch_size=8)
    torch.manual_seed(11)              # IDENTICAL data on every rank
    for _ in range(2):
        x = torch.randn(8, 32)
        y = torch.randint(0, 4, (8,))
        opt.zero_grad()
        torch.nn.functional.cross_entropy(model(x), y).backward()
        opt.step()
    gns = opt.noise_scale
    kf.finalize()
    return gns


def test_gns_zero_when_ranks_agree(port_block):
    """Identical data on all ranks => local grad == averaged grad =>
    the noise-scale estimator's S term is ~0 (formula sanity check,
    reference ops/monitor.py:6-18)."""
    results = spawn_cluster(gns_zero_noise_body, 2, port_block)
    for gns in results:
        assert abs(gns) < 1e-3, gns
