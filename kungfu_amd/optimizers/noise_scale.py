"""Gradient noise scale monitor (McCandlish et al. "An Empirical Model of
Large-Batch Training" estimator).

Reference parity: srcs/python/kungfu/tensorflow/optimizers/grad_noise_scale.py
+ ops/monitor.py:6-18 (the gns formula) + the NoiseScale EMA op
(ops/cpu/collective.cpp:212-258). Wraps S-SGD: before averaging we have the
local (small-batch B) gradient, after averaging the global (big-batch N*B)
gradient; their squared norms give unbiased estimators

    |G|^2_est = (B_big*|G_big|^2 - B_small*|G_small|^2) / (B_big - B_small)
    S_est     = (|G_small|^2 - |G_big|^2) / (1/B_small - 1/B_big)
    gns       = S_est / |G|^2_est        (EMA-smoothed)

Norms are computed on-device by the HIP norm² kernel over the fused grad
buckets — no extra kernel per parameter.
"""
import torch

from kungfu_amd import _core
from kungfu_amd.optimizers.core import KungFuOptimizer
from kungfu_amd.parallel.fusion import GradBucketReducer


class GradNoiseScaleProbe:
    """Standalone gradient-noise-scale monitor usable with ANY optimizer
    (BASELINE config 4 pairs it with SynchronousAveraging, where gradients
    are never all-reduced by the training path): every `interval` steps it
    all-reduces a copy of the gradients to obtain the big-batch gradient
    and updates the EMA estimators."""

    def __init__(self, params, device_batch_size, alpha=0.6, interval=10):
        import kungfu_amd  # noqa: F401 (ensure init)

        self.params = [p for p in params if p.requires_grad]
        self.b_small = float(device_batch_size)
        self.alpha = float(alpha)
        self.interval = max(1, int(interval))
        self._steps = 0
        self._ema_g2 = None
        self._ema_s = None
        self.noise_scale = float("nan")
        self._probe = None

    def _sqnorm(self, tensors):
        if tensors[0].is_cuda:
            from kungfu_amd.ops import hip as hip_ops

            return float(hip_ops.norm2_multi(tensors).item())
        return sum(float(t.float().pow(2).sum().item()) for t in tensors)

    def observe(self):
        """Call after backward() (before the optimizer consumes grads)."""
        from kungfu_amd import _core
        from kungfu_amd.ops import all_reduce

        self._steps += 1
        if _core.size() <= 1 or (self._steps - 1) % self.interval:
            return self.noise_scale
        grads = [p.grad for p in self.params if p.grad is not None]
        if not grads:
            return self.noise_scale
        if self._probe is None:
            total = sum(g.numel() for g in grads)
            self._probe = torch.empty(total, dtype=grads[0].dtype,
                                      device=grads[0].device)
        off = 0
        for g in grads:
            self._probe[off:off + g.numel()].copy_(g.reshape(-1))
            off += g.numel()
        probe = self._probe[:off]
        g_small_sq = self._sqnorm([probe])
        all_reduce(probe, name="|gnsprobe", average=True)
        g_big_sq = self._sqnorm([probe])
        n = _core.size()
        b_small, b_big = self.b_small, self.b_small * n
        g2 = (b_big * g_big_sq - b_small * g_small_sq) / (b_big - b_small)
        s = (g_small_sq - g_big_sq) / (1.0 / b_small - 1.0 / b_big)
        a = self.alpha
        self._ema_g2 = g2 if self._ema_g2 is None else (
            a * g2 + (1 - a) * self._ema_g2)
        self._ema_s = s if self._ema_s is None else (
            a * s + (1 - a) * self._ema_s)
        if self._ema_g2:
            self.noise_scale = self._ema_s / self._ema_g2
        return self.noise_scale


class MonitorGradientNoiseScaleOptimizer(KungFuOptimizer):
    def __init__(self, optimizer, device_batch_size, alpha=0.6,
                 monitor_interval=1):
        super().__init__(optimizer)
        self.b_small = float(device_batch_size)
        self.alpha = float(alpha)
        self.interval = int(monitor_interval)
        self._steps = 0
        self._ema_g2 = None
        self._ema_s = None
        self.noise_scale = float("nan")
        # overlap disabled: we need the pre-average local grad norm
        self.reducer = GradBucketReducer(self._params(), overlap=False)

    def zero_grad(self, set_to_none=False):
        self.reducer.zero_grad()

    def _grad_sqnorm(self):
        flats = [b.flat for b in self.reducer.buckets]
        if flats[0].is_cuda:
            from kungfu_amd.ops import hip as hip_ops

            return float(hip_ops.norm2_multi(flats).item())
        return sum(float(f.float().pow(2).sum().item()) for f in flats)

    def _step(self):
        monitor = (self._steps % self.interval == 0) and _core.size() > 1
        if monitor:
            g_small_sq = self._grad_sqnorm()
        self.reducer.finalize()  # all-reduce + average
        if monitor:
            g_big_sq = self._grad_sqnorm()
            n = _core.size()
            b_small, b_big = self.b_small, self.b_small * n
            if b_big > b_small:
                g2 = (b_big * g_big_sq - b_small * g_small_sq) / (
                    b_big - b_small)
                s = (g_small_sq - g_big_sq) / (1.0 / b_small - 1.0 / b_big)
                a = self.alpha
                self._ema_g2 = g2 if self._ema_g2 is None else (
                    a * g2 + (1 - a) * self._ema_g2)
                self._ema_s = s if self._ema_s is None else (
                    a * s + (1 - a) * self._ema_s)
                if self._ema_g2 != 0:
                    self.noise_scale = self._ema_s / self._ema_g2
        self.optimizer.step()
        self._steps += 1
