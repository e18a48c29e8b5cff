"""Gradient variance monitor.

Reference parity: srcs/python/kungfu/tensorflow/optimizers/grad_variance.py
(:47-59): Var = E[|g_i|^2] - |E[g_i]|^2 across workers, using the local
grad norm (before averaging) and the averaged grad norm (after).
"""
import torch

from kungfu_amd import _core
from kungfu_amd.ops import all_reduce
from kungfu_amd.optimizers.core import KungFuOptimizer
from kungfu_amd.parallel.fusion import GradBucketReducer


class MonitorGradientVarianceOptimizer(KungFuOptimizer):
    def __init__(self, optimizer, monitor_interval=1):
        super().__init__(optimizer)
        self.interval = int(monitor_interval)
        self._steps = 0
        self.variance = float("nan")
        self.reducer = GradBucketReducer(self._params(), overlap=False)

    def zero_grad(self, set_to_none=False):
        self.reducer.zero_grad()

    def _sqnorm(self):
        flats = [b.flat for b in self.reducer.buckets]
        if flats[0].is_cuda:
            from kungfu_amd.ops import hip as hip_ops

            return float(hip_ops.norm2_multi(flats).item())
        return sum(float(f.float().pow(2).sum().item()) for f in flats)

    def _step(self):
        monitor = (self._steps % self.interval == 0) and _core.size() > 1
        if monitor:
            local_sq = torch.tensor([self._sqnorm()], dtype=torch.float64)
            all_reduce(local_sq, name="|gvar", average=True)
            mean_sq = float(local_sq[0])
        self.reducer.finalize()
        if monitor:
            avg_sq = self._sqnorm()
            self.variance = mean_sq - avg_sq
        self.optimizer.step()
        self._steps += 1
