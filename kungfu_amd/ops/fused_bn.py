"""Fused BatchNorm(+residual)(+ReLU) for NHWC bf16 activations.

Wraps the gfx950 kernels in csrc/hip/fused_bn.hip as a torch autograd
Function + module. Used by the ResNet-50 benchmark path instead of
autocast's fp32 BN (which costs bf16<->f32 cast passes + separate
residual-add and ReLU kernels). Parameters and running stats are fp32;
activations stay bf16 end to end.

Requirements for the fused path: 4D channels_last contiguous bf16 input,
C % 8 == 0, C <= 2048. The module falls back to eager torch BN otherwise
(and always on CPU), so numerics tests can compare both paths.
"""
import torch

try:
    from kungfu_amd import _hip
except ImportError:  # pragma: no cover
    _hip = None


def _stream():
    return torch.cuda.current_stream().cuda_stream


def _nhwc_ok(x):
    return (x.dim() == 4 and x.dtype == torch.bfloat16 and
            x.is_contiguous(memory_format=torch.channels_last) and
            x.shape[1] % 8 == 0 and x.shape[1] <= 2048)


class _FusedBNFunction(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, residual, weight, bias, running_mean, running_var,
                momentum, eps, relu, training, ws):
        C = x.shape[1]
        M = x.numel() // C
        s = _stream()
        dev = x.device
        if training:
            # persistent per-module workspace: zero_() only, no per-call
            # allocations (was ~100 torch.zeros launches/step on ResNet-50)
            # fwd_sums is zeroed by the PREVIOUS step's bn_finalize
            # (zero-fused epilogue): no fill launch here
            sums = ws["fwd_sums"]
            _hip.bn_stats(x.data_ptr(), M, C, sums.data_ptr(), s)
            save_mean = ws["save_mean"]
            save_rstd = ws["save_rstd"]
            a = ws["a"]
            b = ws["b"]
            _hip.bn_finalize(sums.data_ptr(), weight.data_ptr(),
                             bias.data_ptr(), running_mean.data_ptr(),
                             running_var.data_ptr(), save_mean.data_ptr(),
                             save_rstd.data_ptr(), a.data_ptr(),
                             b.data_ptr(), M, C, float(eps),
                             float(momentum), s)
        else:
            rstd = torch.rsqrt(running_var + eps)
            a = (weight * rstd).float()
            b = (bias - running_mean * a).float()
            save_mean, save_rstd = running_mean, rstd
        y = torch.empty_like(x)
        mask = None
        if relu and training:
            # 1 bit/elem ReLU sign mask: backward never re-reads the
            # residual or recomputes the pre-activation
            mask = torch.empty(M * (C // 8), dtype=torch.uint8, device=dev)
        _hip.bn_fwd(x.data_ptr(),
                    residual.data_ptr() if residual is not None else 0,
                    y.data_ptr(), a.data_ptr(), b.data_ptr(), M, C,
                    bool(relu), mask.data_ptr() if mask is not None else 0,
                    s)
        ctx.save_for_backward(x, a, save_mean, save_rstd,
                              mask if mask is not None else x)
        ctx.has_res = residual is not None
        ctx.has_mask = mask is not None
        ctx.MC = (M, C)
        ctx.ws = ws
        return y

    @staticmethod
    def backward(ctx, dy):
        x, a, save_mean, save_rstd, mask = ctx.saved_tensors
        M, C = ctx.MC
        s = _stream()
        if dy.dtype != x.dtype:  # kernels require bf16 dy (matching x)
            dy = dy.to(x.dtype)
        dy = dy.contiguous(memory_format=torch.channels_last)
        # bwd_sums was re-zeroed by the previous step's bn_fold
        sums = ctx.ws["bwd_sums"]
        dbdw = ctx.ws["dbdw"]
        mask_ptr = mask.data_ptr() if ctx.has_mask else 0
        _hip.bn_bwd_reduce(dy.data_ptr(), x.data_ptr(), mask_ptr,
                           save_mean.data_ptr(), save_rstd.data_ptr(), M,
                           C, sums.data_ptr(), s)
        # fold shadows into dbdw and re-zero them (no fill next step)
        _hip.bn_fold(sums.data_ptr(), C, dbdw.data_ptr(), s)
        dx = torch.empty_like(x)
        dres = torch.empty_like(x) if ctx.has_res else None
        _hip.bn_bwd_dx(dy.data_ptr(), x.data_ptr(), mask_ptr, a.data_ptr(),
                       save_mean.data_ptr(), save_rstd.data_ptr(),
                       dbdw.data_ptr(), M, C, dx.data_ptr(),
                       dres.data_ptr() if dres is not None else 0, s)
        db = dbdw[:C]               # db = sum(dy_m)
        dw = dbdw[C:2 * C]          # dw = sum(dy_m * xhat)
        return (dx, dres, dw, db, None, None, None, None, None, None,
                None)


class FusedBNReLU2d(torch.nn.Module):
    """Drop-in BatchNorm2d with optional fused ReLU and residual add:
    forward(x, residual=None) = [relu](bn(x) [+ residual])."""

    def __init__(self, channels, eps=1e-5, momentum=0.1, relu=True):
        super().__init__()
        self.channels = channels
        self.eps = eps
        self.momentum = momentum
        self.relu = relu
        self.weight = torch.nn.Parameter(torch.ones(channels))
        self.bias = torch.nn.Parameter(torch.zeros(channels))
        self.register_buffer("running_mean", torch.zeros(channels))
        self.register_buffer("running_var", torch.ones(channels))
        self._ws = None

    def _workspace(self, dev):
        """Persistent kernel workspace (shadow accumulators, folded
        scale/shift, saved stats): fixed addresses keep hipGraph capture
        stable and remove ~4 small allocations+fills per call. Relies on
        the standard one-forward-one-backward pairing per step."""
        if self._ws is None or self._ws["a"].device != dev:
            C = self.channels
            f32 = torch.float32
            self._ws = {
                "fwd_sums": torch.zeros(16 * C, dtype=f32, device=dev),
                "bwd_sums": torch.zeros(16 * C, dtype=f32, device=dev),
                "dbdw": torch.empty(2 * C, dtype=f32, device=dev),
                "save_mean": torch.empty(C, dtype=f32, device=dev),
                "save_rstd": torch.empty(C, dtype=f32, device=dev),
                "a": torch.empty(C, dtype=f32, device=dev),
                "b": torch.empty(C, dtype=f32, device=dev),
            }
        return self._ws

    def _params_f32(self):
        # the kernels reinterpret these pointers as float32: after e.g.
        # model.to(torch.bfloat16) the fused path must fall back to eager
        # instead of reading bf16 bytes as f32 garbage
        return (self.weight.dtype == torch.float32 and
                self.bias.dtype == torch.float32 and
                self.running_mean.dtype == torch.float32 and
                self.running_var.dtype == torch.float32)

    def forward(self, x, residual=None):
        if _hip is not None and x.is_cuda and _nhwc_ok(x) and (
                residual is None or _nhwc_ok(residual)) and \
                self._params_f32():
            return _FusedBNFunction.apply(
                x, residual, self.weight, self.bias, self.running_mean,
                self.running_var, self.momentum, self.eps, self.relu,
                self.training, self._workspace(x.device))
        # eager fallback (CPU, odd shapes): numerically the reference
        y = torch.nn.functional.batch_norm(
            x.float(), self.running_mean, self.running_var, self.weight,
            self.bias, self.training, self.momentum, self.eps)
        if residual is not None:
            y = y + residual.float()
        if self.relu:
            y = torch.nn.functional.relu(y)
        return y.to(x.dtype)
