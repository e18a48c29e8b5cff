"""Single-GPU end-to-end training through the full stack (gpu-marked).

Multi-GPU paths are covered by CPU multi-process tests (gloo-free, via the
C++ engine) and by the driver's round-end scaling bench."""
import pytest
import torch

pytestmark = pytest.mark.gpu


def test_resnet50_bf16_step_runs():
    import kungfu_amd as kf
    from kungfu_amd.models import resnet50
    from kungfu_amd.optimizers import SynchronousSGDOptimizer

    kf.init()
    model = resnet50().to(device="cuda", dtype=torch.bfloat16)
    opt = SynchronousSGDOptimizer(
        torch.optim.SGD(model.parameters(), lr=0.1, momentum=0.9))
    x = torch.randn(8, 3, 224, 224, device="cuda", dtype=torch.bfloat16)
    y = torch.randint(0, 1000, (8,), device="cuda")
    losses = []
    for _ in range(3):
        opt.zero_grad()
        loss = torch.nn.functional.cross_entropy(model(x).float(), y)
        loss.backward()
        opt.step()
        losses.append(float(loss))
    torch.cuda.synchronize()
    assert all(l == l and l < 1e4 for l in losses)  # finite, no NaN/inf


def test_grad_views_survive_backward():
    """param.grad views into fused buckets receive the real gradients."""
    import kungfu_amd as kf
    from kungfu_amd.parallel.fusion import GradBucketReducer

    kf.init()
    lin = torch.nn.Linear(64, 64).to("cuda", torch.float32)
    red = GradBucketReducer(list(lin.parameters()))
    x = torch.randn(4, 64, device="cuda")
    out = lin(x).sum()
    out.backward()
    ref_w_grad = x.sum(0).repeat(64, 1)
    assert torch.allclose(lin.weight.grad, ref_w_grad, atol=1e-4)
    assert lin.weight.grad.data_ptr() >= red.buckets[0].flat.data_ptr()


def test_bert_bf16_step_runs():
    import kungfu_amd as kf
    from kungfu_amd.models import bert_base
    from kungfu_amd.optimizers import MonitorGradientNoiseScaleOptimizer

    kf.init()
    model = bert_base(max_len=128).to(device="cuda", dtype=torch.bfloat16)
    opt = MonitorGradientNoiseScaleOptimizer(
        torch.optim.SGD(model.parameters(), lr=0.01),
        device_batch_size=2)
    ids = torch.randint(0, 30522, (2, 64), device="cuda")
    labels = torch.randint(0, 30522, (2, 64), device="cuda")
    opt.zero_grad()
    out = model(ids)
    loss = torch.nn.functional.cross_entropy(out.float().flatten(0, 1),
                                             labels.flatten())
    loss.backward()
    opt.step()
    torch.cuda.synchronize()
    assert float(loss) == float(loss)


def test_sma_pair_gpu_single():
    import kungfu_amd as kf
    from kungfu_amd.models import SLP
    from kungfu_amd.optimizers import (PairAveragingOptimizer,
                                       SynchronousAveragingOptimizer)

    kf.init()
    for cls in (SynchronousAveragingOptimizer, PairAveragingOptimizer):
        m = SLP().to("cuda", torch.bfloat16)
        opt = cls(torch.optim.SGD(m.parameters(), lr=0.1))
        x = torch.randn(4, 1, 28, 28, device="cuda", dtype=torch.bfloat16)
        y = torch.randint(0, 10, (4,), device="cuda")
        opt.zero_grad()
        torch.nn.functional.cross_entropy(m(x).float(), y).backward()
        opt.step()
    torch.cuda.synchronize()


def test_fused_sgd_step_matches_torch():
    import kungfu_amd as kf
    from kungfu_amd.optimizers import SynchronousSGDOptimizer

    kf.init()
    torch.manual_seed(5)
    lin1 = torch.nn.Sequential(torch.nn.Linear(64, 64),
                               torch.nn.Linear(64, 8)).to("cuda")
    lin2 = torch.nn.Sequential(torch.nn.Linear(64, 64),
                               torch.nn.Linear(64, 8)).to("cuda")
    lin2.load_state_dict(lin1.state_dict())
    opt1 = SynchronousSGDOptimizer(
        torch.optim.SGD(lin1.parameters(), lr=0.1, momentum=0.9,
                        weight_decay=1e-4), fused_step=True)
    opt2 = torch.optim.SGD(lin2.parameters(), lr=0.1, momentum=0.9,
                           weight_decay=1e-4)
    for i in range(3):
        x = torch.randn(16, 64, device="cuda")
        opt1.zero_grad()
        lin1(x).pow(2).mean().backward()
        opt1.step()
        opt2.zero_grad()
        lin2(x).pow(2).mean().backward()
        opt2.step()
    torch.cuda.synchronize()
    for p1, p2 in zip(lin1.parameters(), lin2.parameters()):
        assert torch.allclose(p1, p2, atol=1e-5, rtol=1e-5), \
            (p1 - p2).abs().max()


def test_sma_pair_with_channels_last_model():
    """SMA/Pair fuse channels_last conv weights through the raw-storage
    pack kernels (regression: dense-but-not-row-major layouts)."""
    import kungfu_amd as kf
    from kungfu_amd.models import resnet50
    from kungfu_amd.optimizers import (PairAveragingOptimizer,
                                       SynchronousAveragingOptimizer)

    kf.init()
    for cls in (SynchronousAveragingOptimizer, PairAveragingOptimizer):
        m = resnet50().to("cuda").to(memory_format=torch.channels_last)
        opt = cls(torch.optim.SGD(m.parameters(), lr=0.01))
        x = torch.randn(2, 3, 64, 64, device="cuda").contiguous(
            memory_format=torch.channels_last)
        y = torch.randint(0, 1000, (2,), device="cuda")
        opt.zero_grad()
        with torch.autocast("cuda", dtype=torch.bfloat16):
            out = m(x)
        torch.nn.functional.cross_entropy(out.float(), y).backward()
        opt.step()
    torch.cuda.synchronize()


def test_bf16_master_training_converges():
    """The flagship bench mode (bf16 conv/linear weights + f32 masters in
    the fused optimizer + pack-mode gradients): loss must strictly
    decrease on an overfit-one-batch problem, proving grads flow through
    pack -> flat buckets -> sgd_momentum_master and the bf16 params track
    the f32 masters."""
    import kungfu_amd as kf
    from kungfu_amd.models import resnet50
    from kungfu_amd.optimizers import SynchronousSGDOptimizer
    from kungfu_amd.utils.precision import convert_bf16_master

    kf.init()
    torch.manual_seed(7)
    model = resnet50(fused_bn=True).to("cuda")
    convert_bf16_master(model)
    model = model.to(memory_format=torch.channels_last)
    opt = SynchronousSGDOptimizer(
        torch.optim.SGD(model.parameters(), lr=0.005, momentum=0.9),
        fused_step=True)
    assert opt.reducer.mode == "pack"
    x = torch.randn(8, 3, 224, 224, device="cuda",
                    dtype=torch.bfloat16).contiguous(
        memory_format=torch.channels_last)
    y = torch.randint(0, 1000, (8,), device="cuda")
    losses = []
    for step in range(12):
        opt.zero_grad()
        loss = torch.nn.functional.cross_entropy(model(x).float(), y)
        loss.backward()
        opt.step()
        if step == 0:
            # pack-kernel mechanics: every flat bucket must hold exactly
            # the autograd grads, packed at the bucket offsets
            torch.cuda.synchronize()
            for b in opt.reducer.buckets:
                for p, off in zip(b.params, b.offsets):
                    g = b.flat[off:off + p.numel()]
                    pg = p.grad
                    if pg.dim() == 4 and pg.is_contiguous(
                            memory_format=torch.channels_last):
                        # pack copies MEMORY order (NHWC)
                        ref = pg.permute(0, 2, 3, 1).reshape(-1)
                    else:
                        ref = pg.reshape(-1)
                    assert torch.equal(g, ref), "pack mismatch"
        losses.append(float(loss.detach()))
    torch.cuda.synchronize()
    assert all(l == l for l in losses), losses  # no NaN
    # overfitting one batch at a mild lr: loss must trend down
    assert losses[-1] < losses[0], losses
    # masters and bf16 params stay in sync
    b = opt.reducer.buckets[0]
    if b.master is not None:
        assert torch.equal(b.param_flat,
                           b.master.to(torch.bfloat16))


def test_checkpoint_roundtrip_fused_master(tmp_path):
    """Checkpoint/restore through the fused-master optimizer: f32 masters
    and momentum live in the fused flat buffers, and restored training
    must continue bit-identically."""
    import kungfu_amd as kf
    from kungfu_amd.models import SLP
    from kungfu_amd.optimizers import SynchronousSGDOptimizer
    from kungfu_amd.parallel.elastic import (load_checkpoint,
                                             save_checkpoint)
    from kungfu_amd.utils.precision import convert_bf16_master

    kf.init()
    torch.manual_seed(6)

    def make():
        m = convert_bf16_master(SLP(in_features=8, classes=3).to("cuda"))
        o = SynchronousSGDOptimizer(
            torch.optim.SGD(m.parameters(), lr=0.1, momentum=0.9),
            fused_step=True)
        return m, o

    model, opt = make()
    x = torch.randn(4, 1, 2, 4, device="cuda", dtype=torch.bfloat16)
    y = torch.randint(0, 3, (4,), device="cuda")
    for _ in range(3):
        opt.zero_grad()
        torch.nn.functional.cross_entropy(model(x).float(), y).backward()
        opt.step()
    torch.cuda.synchronize()
    path = str(tmp_path / "ck")
    save_checkpoint(path, model, opt, step=3)
    model2, opt2 = make()
    step, _ = load_checkpoint(path, model2, opt2, map_location="cuda")
    assert step == 3
    for b1, b2 in zip(opt.reducer.buckets, opt2.reducer.buckets):
        if b1.master is not None:
            assert torch.equal(b1.master, b2.master)
        assert torch.equal(b1.momentum, b2.momentum)
    for m, o in ((model, opt), (model2, opt2)):
        o.zero_grad()
        torch.nn.functional.cross_entropy(m(x).float(), y).backward()
        o.step()
    torch.cuda.synchronize()
    for b1, b2 in zip(opt.reducer.buckets, opt2.reducer.buckets):
        assert torch.equal(b1.param_flat, b2.param_flat)


def test_pack_mode_gradient_accumulation():
    """overlap=False + several backwards per step (gradient accumulation)
    in pack mode: the packed flats must hold the ACCUMULATED autograd
    grads (regression: finalize once launched without packing)."""
    import kungfu_amd as kf
    from kungfu_amd.optimizers import SynchronousSGDOptimizer

    kf.init()
    torch.manual_seed(9)
    lin = torch.nn.Linear(32, 16).to("cuda", torch.bfloat16)
    opt = SynchronousSGDOptimizer(
        torch.optim.SGD(lin.parameters(), lr=0.0, momentum=0.0),
        overlap=False, fused_step=True)
    assert opt.reducer.mode == "pack"
    opt.zero_grad()
    xs = [torch.randn(8, 32, device="cuda", dtype=torch.bfloat16)
          for _ in range(3)]
    for x in xs:
        lin(x).sum().backward()
    accumulated = lin.weight.grad.clone()
    opt.step()  # lr=0: packs + (world-1) fused apply, params unchanged
    torch.cuda.synchronize()
    b = opt.reducer.bucket_of[lin.weight]
    idx = next(i for i, p in enumerate(b.params) if p is lin.weight)
    off = b.offsets[idx]
    flat = b.flat[off:off + lin.weight.numel()]
    assert torch.equal(flat, accumulated.reshape(-1))
