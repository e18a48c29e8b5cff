"""Synchronous S-SGD: all-reduce gradients, average, apply locally.

Reference parity: srcs/python/kungfu/tensorflow/optimizers/sync_sgd.py and
kungfu/torch/optimizers/sync_sgd.py. MI355X-native: gradients live in fused
flat buckets (grad-as-view), all-reduced over RCCL/xGMI from backward hooks
with deterministic bucket order (see kungfu_amd/parallel/fusion.py); on CPU
clusters the C++ graph-strategy engine carries the reduction.
"""
import torch

from kungfu_amd.optimizers.core import KungFuOptimizer
from kungfu_amd.parallel.fusion import (DEFAULT_BUCKET_BYTES, _ALIGN,
                                        GradBucketReducer, _alias_view)


class SynchronousSGDOptimizer(KungFuOptimizer):
    def __init__(self, optimizer, bucket_bytes=DEFAULT_BUCKET_BYTES,
                 overlap=True, average=True, name="sgd", fused_step=False):
        super().__init__(optimizer)
        # fused step -> pack-mode gradients: autograd keeps its own grad
        # tensors (first accumulation is a steal, no per-param zero/add
        # kernels) and one HIP pack per bucket feeds the flat buffers the
        # fused optimizer consumes
        self.reducer = GradBucketReducer(self._params(),
                                         bucket_bytes=bucket_bytes,
                                         average=average, overlap=overlap,
                                         name=name,
                                         mode="pack" if fused_step
                                         else "alias")
        self.fused_step = fused_step
        if fused_step:
            self._build_fused_step()

    def zero_grad(self, set_to_none=False):
        # grads are views into the fused buckets: always zero in place
        self.reducer.zero_grad()

    # -- fused SGD over flat buckets (gfx950 kernel) -------------------

    def _build_fused_step(self):
        """Alias parameter storage onto flat buffers with the SAME layout
        as the grad buckets, so the whole SGD-momentum step is one HIP
        kernel per bucket (vs ~5 foreach launches per tensor group), with
        the 1/N gradient average folded into the kernel's grad_scale."""
        groups = self.optimizer.param_groups
        if len(groups) != 1:
            raise ValueError("fused_step supports a single param group")
        g = groups[0]
        if g.get("nesterov") and g.get("momentum", 0) == 0:
            raise ValueError("nesterov requires momentum")
        for b in self.reducer.buckets:
            flat = torch.zeros_like(b.flat)
            mom = torch.zeros(b.flat.numel(), dtype=torch.float32,
                              device=b.flat.device)
            off = 0
            for p in b.params:
                view = _alias_view(flat[off:off + p.numel()], p)
                view.copy_(p.data)
                p.data = view
                off = (off + p.numel() + _ALIGN - 1) // _ALIGN * _ALIGN
            b.param_flat = flat
            b.momentum = mom
            # low-precision buckets (bf16-master mode): the authoritative
            # f32 copy lives here and the fused kernel keeps param/master
            # in sync — no separate cast passes (vs autocast's per-step
            # bf16<->f32 master-weight copies)
            b.master = (flat.to(torch.float32)
                        if flat.dtype != torch.float32 else None)

    def _fused_apply(self):
        from kungfu_amd.ops import hip as hip_ops

        g = self.optimizer.param_groups[0]
        scale = self.reducer.grad_scale
        for b in self.reducer.buckets:
            if b.master is not None:
                hip_ops.sgd_momentum_master(
                    b.param_flat, b.flat, b.master, b.momentum, lr=g["lr"],
                    momentum=g.get("momentum", 0.0),
                    weight_decay=g.get("weight_decay", 0.0),
                    grad_scale=scale, nesterov=g.get("nesterov", False))
            else:
                hip_ops.sgd_momentum(
                    b.param_flat, b.flat, b.momentum, lr=g["lr"],
                    momentum=g.get("momentum", 0.0),
                    weight_decay=g.get("weight_decay", 0.0),
                    grad_scale=scale, nesterov=g.get("nesterov", False))

    def state_dict(self):
        sd = {"inner": self.optimizer.state_dict()}
        if self.fused_step:
            # momentum lives in the fused flat buffers, not in torch state
            sd["fused_momentum"] = [b.momentum for b in
                                    self.reducer.buckets]
            sd["fused_master"] = [b.master for b in self.reducer.buckets]
        return sd

    def load_state_dict(self, sd):
        if "inner" not in sd:  # plain torch-style dict (older checkpoints)
            self.optimizer.load_state_dict(sd)
            return
        self.optimizer.load_state_dict(sd["inner"])
        if self.fused_step and "fused_momentum" in sd:
            for b, m in zip(self.reducer.buckets, sd["fused_momentum"]):
                b.momentum.copy_(m.to(b.momentum.device))
        if self.fused_step and "fused_master" in sd:
            for b, m in zip(self.reducer.buckets, sd["fused_master"]):
                if b.master is not None and m is not None:
                    b.master.copy_(m.to(b.master.device))

    def _step(self):
        if self.fused_step:
            # skip the separate grad-average pass: 1/N is folded into the
            # fused kernel's grad_scale
            avg = self.reducer.average
            self.reducer.average = False
            self.reducer.finalize()
            self.reducer.average = avg
            self._fused_apply()
            return
        self.reducer.finalize()
        self.optimizer.step()
