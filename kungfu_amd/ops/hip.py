"""Torch wrappers over the hand-written gfx950 HIP kernels.

The kernels live in csrc/hip/kernels.hip (pack/unpack fusion, fused model
averaging, norm² reduction, fused SGD-momentum). On a GPU machine the
extension MUST be present — these ops raise instead of silently falling
back to eager torch, so a missing native build is loud.
"""
import torch

try:
    from kungfu_amd import _hip
except ImportError:  # pragma: no cover - CPU-only environments
    _hip = None

from kungfu_amd.utils.dtypes import hip_dtype


def available():
    return _hip is not None


def _require():
    if _hip is None:
        raise RuntimeError(
            "kungfu_amd._hip is not built; run `python setup.py build_ext "
            "--inplace` (gfx950 HIP kernels are mandatory on GPU hosts)")
    return _hip


def _stream():
    return torch.cuda.current_stream().cuda_stream


def _dense(t):
    # dense storage in SOME layout (row-major or channels_last): raw
    # pointer kernels only need the numel elements to be one dense block
    if t.is_contiguous():
        return True
    return t.dim() == 4 and t.is_contiguous(
        memory_format=torch.channels_last)


def _check_cuda(*tensors):
    for t in tensors:
        if not t.is_cuda:
            raise ValueError("HIP kernel requires CUDA(HIP) tensors")
        if not _dense(t):
            raise ValueError("HIP kernel requires dense tensors")


class FusionPlan:
    """Device-side chunk table for packing a list of same-dtype tensors into
    (or out of) a flat fused buffer.

    Holds references to the tensors (keeping their storage alive) and
    revalidates their data_ptrs on every pack/unpack: `p.data` reassignment
    (fused-step flat aliasing, elastic rebuilds, model.to(...)) would
    otherwise leave the device-side chunk table pointing at freed or
    unrelated memory. On mismatch the chunk table is rebuilt in place."""

    def __init__(self, tensors, offsets, dtype, chunk_elems=1 << 14):
        _require()
        _check_cuda(*tensors)
        self._tensors = list(tensors)
        self._offsets = [int(o) for o in offsets]
        self._chunk_elems = chunk_elems
        self._ptrs = [t.data_ptr() for t in self._tensors]
        self._plan = self._build()
        self.dtype = dtype

    def _build(self):
        segs = [(t.data_ptr(), off, t.numel())
                for t, off in zip(self._tensors, self._offsets)]
        return _hip.FusionPlan(segs, hip_dtype(self._tensors[0].dtype),
                               self._chunk_elems)

    def _revalidate(self):
        ptrs = [t.data_ptr() for t in self._tensors]
        if ptrs != self._ptrs:
            self._ptrs = ptrs
            segs = [(t.data_ptr(), off, t.numel())
                    for t, off in zip(self._tensors, self._offsets)]
            # async table re-point (pinned staging): safe per-step cost
            self._plan.update(segs, _stream())

    def set_sources(self, tensors):
        """Re-point the plan at new source tensors with the same segment
        layout (pack-mode gradients: autograd allocates fresh grad
        tensors every step)."""
        self._tensors = list(tensors)
        self._revalidate()

    def pack(self, fused):
        _check_cuda(fused)
        self._revalidate()
        self._plan.pack(fused.data_ptr(), _stream())

    def unpack(self, fused, scale=1.0):
        _check_cuda(fused)
        self._revalidate()
        self._plan.unpack(fused.data_ptr(), float(scale), _stream())

    @property
    def total_elems(self):
        return self._plan.total_elems

    @property
    def nchunks(self):
        return self._plan.nchunks


def avg_inplace(y, x, alpha=0.5):
    """y <- (1-alpha)*y + alpha*x (model averaging)."""
    _require()
    _check_cuda(y, x)
    assert y.numel() == x.numel() and y.dtype == x.dtype
    _hip.avg_inplace(y.data_ptr(), x.data_ptr(), float(alpha), y.numel(),
                     hip_dtype(y.dtype), _stream())
    return y


def scale_(y, s):
    _require()
    _check_cuda(y)
    _hip.scale_(y.data_ptr(), float(s), y.numel(), hip_dtype(y.dtype),
                _stream())
    return y


def norm2(x, out=None):
    """Sum of squares of x accumulated into a f32 scalar tensor on device."""
    _require()
    _check_cuda(x)
    if out is None:
        out = torch.zeros(1, dtype=torch.float32, device=x.device)
    else:
        out.zero_()
    _hip.norm2(x.data_ptr(), x.numel(), out.data_ptr(), hip_dtype(x.dtype),
               _stream())
    return out


def norm2_multi(tensors, out=None):
    """Accumulated sum of squares over a list of device tensors."""
    h = _require()
    tensors = list(tensors)
    _check_cuda(*tensors)
    if out is None:
        out = torch.zeros(1, dtype=torch.float32,
                          device=tensors[0].device)
    else:
        out.zero_()
    s = _stream()
    for t in tensors:
        h.norm2(t.data_ptr(), t.numel(), out.data_ptr(),
                hip_dtype(t.dtype), s)
    return out


def dot(x, y, out=None):
    _require()
    _check_cuda(x, y)
    assert x.numel() == y.numel() and x.dtype == y.dtype
    if out is None:
        out = torch.zeros(1, dtype=torch.float32, device=x.device)
    else:
        out.zero_()
    _hip.dot(x.data_ptr(), y.data_ptr(), x.numel(), out.data_ptr(),
             hip_dtype(x.dtype), _stream())
    return out


def sgd_momentum(param, grad, momentum_buf, lr, momentum=0.9,
                 weight_decay=0.0, grad_scale=1.0, nesterov=False):
    """Fused SGD step over flat buffers; momentum state is f32."""
    _require()
    _check_cuda(param, grad, momentum_buf)
    assert param.numel() == grad.numel() == momentum_buf.numel()
    assert momentum_buf.dtype == torch.float32
    _hip.sgd_momentum(param.data_ptr(), grad.data_ptr(),
                      momentum_buf.data_ptr(), param.numel(), float(lr),
                      float(momentum), float(weight_decay),
                      float(grad_scale), bool(nesterov),
                      hip_dtype(param.dtype), _stream())


def sgd_momentum_master(param, grad, master, momentum_buf, lr, momentum=0.9,
                        weight_decay=0.0, grad_scale=1.0, nesterov=False):
    """Mixed-precision fused SGD: bf16 params/grads, f32 master + momentum.
    Replaces the autocast master-weight pattern's per-step cast passes."""
    _require()
    _check_cuda(param, grad, master, momentum_buf)
    assert param.numel() == grad.numel() == master.numel() \
        == momentum_buf.numel()
    assert master.dtype == torch.float32
    assert momentum_buf.dtype == torch.float32
    _hip.sgd_momentum_master(param.data_ptr(), grad.data_ptr(),
                             master.data_ptr(), momentum_buf.data_ptr(),
                             param.numel(), float(lr), float(momentum),
                             float(weight_decay), float(grad_scale),
                             bool(nesterov), hip_dtype(param.dtype),
                             _stream())


def transform2(z, x, op="sum"):
    """z <- z op x elementwise on device."""
    _require()
    _check_cuda(z, x)
    ops = {"sum": 0, "min": 1, "max": 2, "prod": 3}
    _hip.transform2(z.data_ptr(), x.data_ptr(), z.numel(), ops[op],
                    hip_dtype(z.dtype), _stream())
    return z
