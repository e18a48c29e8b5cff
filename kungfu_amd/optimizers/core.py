"""Optimizer wrapper base.

Reference parity: srcs/python/kungfu/tensorflow/optimizers/core.py (the
KungFuTFOptimizer/_KungFuAlgorithm split) and kungfu/torch/optimizers/
sync_sgd.py (torch wrapper pattern). Here: a wrapper object that delegates
the full torch.optim.Optimizer surface to the inner optimizer and lets the
algorithm hook before/after the inner step.
"""
import torch


class KungFuOptimizer:
    """Wraps a torch.optim.Optimizer; subclasses implement _step()."""

    def __init__(self, optimizer):
        if not isinstance(optimizer, torch.optim.Optimizer):
            raise TypeError("expected a torch.optim.Optimizer")
        self.optimizer = optimizer

    # --- delegation ---
    @property
    def param_groups(self):
        return self.optimizer.param_groups

    @property
    def defaults(self):
        return self.optimizer.defaults

    @property
    def state(self):
        return self.optimizer.state

    def state_dict(self):
        return self.optimizer.state_dict()

    def load_state_dict(self, sd):
        self.optimizer.load_state_dict(sd)

    def add_param_group(self, g):
        self.optimizer.add_param_group(g)

    def zero_grad(self, set_to_none=False):
        # set_to_none=False default: subclasses may alias grads to fused
        # buffers which must stay allocated
        self.optimizer.zero_grad(set_to_none=set_to_none)

    def _params(self):
        out = []
        for g in self.optimizer.param_groups:
            out.extend(p for p in g["params"] if p.requires_grad)
        return out

    def step(self, closure=None):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()
        self._step()
        return loss

    def _step(self):
        raise NotImplementedError

    def __repr__(self):
        return "%s(%r)" % (type(self).__name__, self.optimizer)
