"""Adaptive SMA -> S-SGD switch.

Reference parity: srcs/python/kungfu/tensorflow/optimizers/ada_sgd.py: run
model averaging for the first `change_step` steps (robust to heterogeneity
early on), then switch to synchronous gradient all-reduce, re-broadcasting
weights from rank 0 at the switch point (AdaSGDHook semantics).
"""
from kungfu_amd.ops import broadcast_parameters
from kungfu_amd.optimizers.core import KungFuOptimizer
from kungfu_amd.optimizers.sma_sgd import SynchronousAveragingOptimizer
from kungfu_amd.optimizers.sync_sgd import SynchronousSGDOptimizer


class AdaptiveSGDOptimizer(KungFuOptimizer):
    def __init__(self, optimizer, change_step, alpha=0.1):
        super().__init__(optimizer)
        self.change_step = int(change_step)
        self._steps = 0
        self._sma = SynchronousAveragingOptimizer(optimizer, alpha=alpha)
        self._sync = None  # built lazily at the switch

    @property
    def synced(self):
        return self._sync is not None

    def zero_grad(self, set_to_none=False):
        if self._sync is not None:
            self._sync.zero_grad()
        else:
            self.optimizer.zero_grad(set_to_none=False)

    def _step(self):
        if self._steps == self.change_step and self._sync is None:
            # switch: resync replicas then move to S-SGD
            broadcast_parameters([p.data for p in self._params()])
            self._sync = SynchronousSGDOptimizer(self.optimizer)
        if self._sync is not None:
            self._sync._step()
        else:
            self._sma._step()
        self._steps += 1
