"""Elastic training helper: step-schedule resize + state resynchronization.

Reference parity: kungfu/tensorflow/hooks/elastic.py (KungFuElasticTrainHook)
+ the StepBasedSchedule op (ops/cpu/elastic.cpp:16-82). Torch-native
version: call `ElasticTrainer.after_step()` once per step; it proposes
scheduled sizes from rank 0, runs resize on every worker, and after a
membership change max-syncs the step counter and re-broadcasts the model
(and freshly rebuilds the optimizer's reducer so grad buckets match the
new cluster).
"""
import torch

import kungfu_amd as kf
from kungfu_amd.ops import broadcast_model


def parse_schedule(spec):
    """'3:2,6:4' -> {3: 2, 6: 4} (at step N resize to size M)."""
    out = {}
    if spec:
        for part in spec.split(","):
            step, size = part.split(":")
            out[int(step)] = int(size)
    return out


class ElasticTrainer:
    def __init__(self, model, make_optimizer, schedule=None):
        """make_optimizer: callable(model) -> KungFu optimizer (rebuilt
        after each resize so fused grad buckets match)."""
        self.model = model
        self.make_optimizer = make_optimizer
        self.schedule = (parse_schedule(schedule)
                         if isinstance(schedule, str) else (schedule or {}))
        self.step = kf.all_reduce_int_max(0)
        broadcast_model(self.model)
        self.optimizer = make_optimizer(model)
        self.detached = False

    def after_step(self):
        """Advance the step; apply any scheduled resize. Returns False when
        this worker was detached and should stop."""
        self.step += 1
        if kf.rank() == 0 and self.step in self.schedule:
            kf.propose_new_size(self.schedule[self.step])
        changed, detached = kf.resize()
        if detached:
            self.detached = True
            return False
        if changed:
            self.step = kf.all_reduce_int_max(self.step)
            broadcast_model(self.model)
            self.optimizer = self.make_optimizer(self.model)
        return True


def save_checkpoint(path, model, optimizer=None, step=0, extra=None):
    """Rank-aware checkpoint save (failure-recovery pattern: reference
    Failure_recovery_examples save per-epoch and reload with --restart)."""
    state = {
        "model": model.state_dict(),
        "step": step,
        "extra": extra or {},
    }
    if optimizer is not None:
        # wrappers define state_dict (covering fused momentum buffers)
        state["optimizer"] = optimizer.state_dict()
    torch.save(state, "%s.rank%d" % (path, kf.rank()))


def load_checkpoint(path, model, optimizer=None, map_location="cpu"):
    """Load this rank's checkpoint; returns (step, extra) or (0, {})."""
    import os

    f = "%s.rank%d" % (path, kf.rank())
    if not os.path.exists(f):
        return 0, {}
    state = torch.load(f, map_location=map_location)
    model.load_state_dict(state["model"])
    if optimizer is not None and "optimizer" in state:
        optimizer.load_state_dict(state["optimizer"])
    return state.get("step", 0), state.get("extra", {})
