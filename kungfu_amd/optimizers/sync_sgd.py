"""Synchronous S-SGD: all-reduce gradients, average, apply locally.

Reference parity: srcs/python/kungfu/tensorflow/optimizers/sync_sgd.py and
kungfu/torch/optimizers/sync_sgd.py. MI355X-native: gradients live in fused
flat buckets (grad-as-view), all-reduced over RCCL/xGMI from backward hooks
with deterministic bucket order (see kungfu_amd/parallel/fusion.py); on CPU
clusters the C++ graph-strategy engine carries the reduction.
"""
from kungfu_amd.optimizers.core import KungFuOptimizer
from kungfu_amd.parallel.fusion import (DEFAULT_BUCKET_BYTES,
                                        GradBucketReducer)


class SynchronousSGDOptimizer(KungFuOptimizer):
    def __init__(self, optimizer, bucket_bytes=DEFAULT_BUCKET_BYTES,
                 overlap=True, average=True, name="sgd"):
        super().__init__(optimizer)
        self.reducer = GradBucketReducer(self._params(),
                                         bucket_bytes=bucket_bytes,
                                         average=average, overlap=overlap,
                                         name=name)

    def zero_grad(self, set_to_none=False):
        # grads are views into the fused buckets: always zero in place
        self.reducer.zero_grad()

    def _step(self):
        self.reducer.finalize()
        self.optimizer.step()
