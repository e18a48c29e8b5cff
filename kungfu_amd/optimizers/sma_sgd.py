"""Synchronous model averaging (SMA / EA-SGD).

Reference parity: srcs/python/kungfu/tensorflow/optimizers/sma_sgd.py:
all-reduce the *weights*, blend v <- (1-alpha)*v + alpha*avg, then apply
local gradients. MI355X-native: the model is packed into one flat buffer by
the HIP fusion kernel, averaged over RCCL, and blended on-device by the
fused averaging kernel (csrc/hip/kernels.hip avg_inplace).
"""
import torch

from kungfu_amd import _core
from kungfu_amd.ops import all_reduce
from kungfu_amd.optimizers.core import KungFuOptimizer
from kungfu_amd.parallel.fusion import FlatParamGroup


class SynchronousAveragingOptimizer(KungFuOptimizer):
    def __init__(self, optimizer, alpha=0.1, name="sma"):
        super().__init__(optimizer)
        self.alpha = float(alpha)
        self.name = name
        self._group = FlatParamGroup(self._params())
        self._is_cuda = self._group.device.type == "cuda"

    @torch.no_grad()
    def _average_weights(self):
        if _core.size() == 1:
            return
        g = self._group
        g.pack()  # flat <- v
        if self._is_cuda:
            from kungfu_amd.ops import hip as hip_ops

            local = g.flat.clone()
            all_reduce(g.flat, name=self.name + "/w")
            g.flat.div_(_core.size())  # flat = avg
            # flat <- (1-(1-alpha))*avg + (1-alpha)*v = (1-a)v + a*avg
            hip_ops.avg_inplace(g.flat, local, alpha=1.0 - self.alpha)
        else:
            local = g.flat.clone()
            all_reduce(g.flat, name=self.name + "/w", average=True)
            g.flat.mul_(self.alpha).add_(local, alpha=1.0 - self.alpha)
        g.unpack()

    def _step(self):
        self._average_weights()
        self.optimizer.step()
