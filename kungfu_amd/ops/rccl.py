"""Native RCCL collective layer — torch-tensor front end.

This is the MI355X-native GPU collective stack (csrc/rccl/): RCCL
communicators bootstrapped over the framework's OWN control plane (the
uniqueId is broadcast through the C++ TCP/Unix collective engine, matching
the reference's nccl bootstrap, srcs/cpp/src/nccl/gpu_collective.cpp:169-191
— no TCPStore, no env:// rendezvous beside the control plane), per-scope
ordered dispatchers (reference scheduler.cpp), and stream-ordered
completion via HIP events.

Scopes (reference controller.cpp): GLOBAL (all ranks), LOCAL (ranks on
this host), CROSS (one master per host) — the latter two power
hierarchical all-reduce without any torch.distributed sub-groups.

On a GPU host the native layer is the DEFAULT GPU path; torch.distributed
is only used when KUNGFU_GPU_BACKEND=torch is exported (fallback/debug).
"""
import torch

from kungfu_amd import _core
from kungfu_amd.utils.dtypes import core_dtype, core_op

try:
    from kungfu_amd import _rccl
except ImportError:  # pragma: no cover - extension not built
    _rccl = None

GLOBAL, LOCAL, CROSS = 0, 1, 2

_state = {"active": False, "cpu": False, "scopes": set()}


def available():
    return _rccl is not None


def active():
    return _state["active"]


def _require():
    if _rccl is None:
        raise RuntimeError(
            "kungfu_amd._rccl is not built; run `python setup.py build_ext"
            " --inplace` (the native RCCL layer is mandatory on GPU hosts)")
    return _rccl


def init_gpu(device=None):
    """Bootstrap the GLOBAL communicator over the control plane."""
    r = _require()
    if _state["active"]:
        return
    if device is None:
        device = torch.cuda.current_device()
    r.init(_core.control_api(), int(device))
    _state["active"] = True
    _state["scopes"] = {GLOBAL}


def init_cpu():
    """Dispatcher-only mode for ordering tests (no GPU)."""
    r = _require()
    r.init_cpu(_core.control_api())
    _state["cpu"] = True
    _state["scopes"] = {GLOBAL, LOCAL, CROSS}


def _ensure_scope(scope):
    r = _require()
    if scope not in _state["scopes"]:
        r.init_scope(scope)
        _state["scopes"].add(scope)


def reinit():
    """Rebuild all live communicators after an elastic resize (reference
    ResetNcclHelper semantics — no process-group destroy/recreate)."""
    if _state["active"] or _state["cpu"]:
        _require().reinit()


def finalize():
    if _rccl is not None and (_state["active"] or _state["cpu"]):
        _rccl.finalize()
    _state["active"] = False
    _state["cpu"] = False
    _state["scopes"] = set()


def _stream():
    return torch.cuda.current_stream().cuda_stream


def _check(t):
    if not t.is_cuda:
        raise ValueError("native RCCL ops require CUDA(HIP) tensors")
    if not (t.is_contiguous() or
            (t.dim() == 4 and
             t.is_contiguous(memory_format=torch.channels_last))):
        raise ValueError("native RCCL ops require dense tensors")


# ---- async API (returns handle ids; wait on a stream or the host) ----

def all_reduce_async(tensor, op="sum", name="", scope=GLOBAL):
    _check(tensor)
    _ensure_scope(scope)
    if torch.cuda.is_current_stream_capturing():
        # hipGraph capture: launch inline on the captured stream (the
        # dispatcher's cross-thread events cannot be captured; capture is
        # single-threaded so program order IS the cross-rank agreement)
        _rccl.all_reduce_inline(scope, tensor.data_ptr(),
                                tensor.data_ptr(), tensor.numel(),
                                core_dtype(tensor.dtype), core_op(op),
                                _stream())
        return 0  # stream-ordered; wait()/wait_host() are no-ops
    return _rccl.all_reduce(scope, name, tensor.data_ptr(),
                            tensor.data_ptr(), tensor.numel(),
                            core_dtype(tensor.dtype), core_op(op),
                            _stream())


def broadcast_async(tensor, root=0, name="", scope=GLOBAL):
    _check(tensor)
    _ensure_scope(scope)
    return _rccl.broadcast(scope, name, tensor.data_ptr(),
                           tensor.data_ptr(), tensor.numel(),
                           core_dtype(tensor.dtype), int(root), _stream())


def reduce_async(tensor, op="sum", root=0, name="", scope=GLOBAL):
    _check(tensor)
    _ensure_scope(scope)
    return _rccl.reduce(scope, name, tensor.data_ptr(), tensor.data_ptr(),
                        tensor.numel(), core_dtype(tensor.dtype),
                        core_op(op), int(root), _stream())


def all_gather_async(out, tensor, name="", scope=GLOBAL):
    _check(tensor)
    _check(out)
    _ensure_scope(scope)
    return _rccl.all_gather(scope, name, tensor.data_ptr(), out.data_ptr(),
                            tensor.numel(), core_dtype(tensor.dtype),
                            _stream())


def reduce_scatter_async(out, tensor, op="sum", name="", scope=GLOBAL):
    _check(tensor)
    _check(out)
    _ensure_scope(scope)
    return _rccl.reduce_scatter(scope, name, tensor.data_ptr(),
                                out.data_ptr(), out.numel(),
                                core_dtype(out.dtype), core_op(op),
                                _stream())


def send_recv_async(send, recv, peer, name="", scope=GLOBAL):
    _check(send)
    _check(recv)
    _ensure_scope(scope)
    assert send.numel() == recv.numel() and send.dtype == recv.dtype
    return _rccl.send_recv(scope, name, send.data_ptr(), recv.data_ptr(),
                           send.numel(), core_dtype(send.dtype), int(peer),
                           _stream())


def wait(handle, stream=None):
    """Order the (current) stream after the collective — no host block."""
    if handle == 0:
        return  # inline (captured) op: already stream-ordered
    _rccl.wait(handle, stream if stream is not None else _stream())


def wait_host(handle):
    if handle == 0:
        return
    _rccl.wait_host(handle)


# ---- sync wrappers (stream-ordered: return once the current stream is
# ordered after the collective; no host sync) ----

def all_reduce(tensor, op="sum", average=False, name=""):
    wait(all_reduce_async(tensor, op=op, name=name))
    if average:
        tensor.div_(scope_size(GLOBAL))
    return tensor


def broadcast(tensor, root=0, name=""):
    wait(broadcast_async(tensor, root=root, name=name))
    return tensor


def reduce(tensor, op="sum", root=0, name=""):
    wait(reduce_async(tensor, op=op, root=root, name=name))
    return tensor


def all_gather(tensor, name=""):
    n = scope_size(GLOBAL)
    out = torch.empty((n,) + tuple(tensor.shape), dtype=tensor.dtype,
                      device=tensor.device)
    wait(all_gather_async(out.view(n, -1), tensor.contiguous().view(-1),
                          name=name))
    return out


def hierarchical_all_reduce(tensor, name=""):
    """local reduce -> cross all-reduce (masters, over xGMI/IB) -> local
    broadcast, all on native sub-communicators (reference
    ops/gpu/collective.cpp:106-158)."""
    _ensure_scope(LOCAL)
    _ensure_scope(CROSS)
    wait(reduce_async(tensor, root=0, name=name, scope=LOCAL))
    if _rccl.scope_member(CROSS):
        wait(all_reduce_async(tensor, name=name, scope=CROSS))
    wait(broadcast_async(tensor, root=0, name=name, scope=LOCAL))
    return tensor


def scope_size(scope):
    return _rccl.scope_size(scope)


def scope_rank(scope):
    return _rccl.scope_rank(scope)


def scope_member(scope):
    _ensure_scope(scope)
    return _rccl.scope_member(scope)


# ---- deterministic ordering surface (reference NCCLScheduler) ----

def scheduler_reset(names, scope=GLOBAL):
    _ensure_scope(scope)
    _rccl.scheduler_reset(scope, list(names))


def scheduler_agree(scope=GLOBAL):
    return list(_rccl.scheduler_agree(scope))


def last_arrival(scope=GLOBAL):
    return list(_rccl.last_arrival(scope))
