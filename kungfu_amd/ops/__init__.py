"""Tensor-level collective ops.

Reference parity: srcs/python/kungfu/tensorflow/ops/ and kungfu/torch/ops/.
Routing (MI355X-native):
  * CUDA tensors -> the NATIVE RCCL layer over xGMI (kungfu_amd.ops.rccl:
    communicators bootstrapped over the own control plane, stream-ordered
    handles); torch.distributed only as the KUNGFU_GPU_BACKEND=torch
    fallback;
  * CPU tensors  -> the C++ collective engine (graph strategies over
    TCP/Unix sockets), zero-copy via data_ptr;
  * `cpu_staged_*` variants stage GPU tensors through host memory into the
    C++ engine — the reference's only torch GPU path
    (src/torch/ops/cuda/collective.cpp:31-54) kept for the cross-host hop
    and plumbing tests.
"""
import collections
import itertools

import torch

from kungfu_amd import _core, _ensure_init
from kungfu_amd.utils.dtypes import core_dtype, core_op

# Per-(kind, numel) sequence counters for auto-derived collective names:
# two successive collectives on distinct same-sized tensors must not share
# a rendezvous name or their payloads could interleave on the CPU engine.
# Counters advance identically on every rank because collectives are (by
# contract) called in the same program order on all ranks. Concurrent
# collectives from MULTIPLE threads must pass explicit unique `name`s.
_name_seq = collections.defaultdict(itertools.count)


def _auto_name(kind, numel):
    return "%s%d#%d" % (kind, numel, next(_name_seq[(kind, numel)]))


def _native():
    """The native RCCL layer module when it is the active GPU backend
    (default on GPU hosts), else None (torch.distributed fallback)."""
    from kungfu_amd.ops import rccl as _r

    return _r if _r.active() else None


class _NativeWork:
    """torch.distributed.Work-alike over a native RCCL handle: wait()
    orders the current stream after the collective (no host block)."""

    __slots__ = ("_handle",)

    def __init__(self, handle):
        self._handle = handle

    def wait(self):
        if self._handle is not None:
            from kungfu_amd.ops import rccl as _r

            _r.wait(self._handle)
            self._handle = None
        return True


def _dist():
    import torch.distributed as dist

    if not dist.is_initialized():
        raise RuntimeError(
            "CUDA collective requested but no GPU backend is up (native "
            "RCCL inactive, torch.distributed uninitialized); call "
            "kungfu_amd.init() first (no silent CPU fallback)")
    return dist


_TORCH_OPS = None


def _torch_reduce_op(op):
    global _TORCH_OPS
    import torch.distributed as dist

    if _TORCH_OPS is None:
        _TORCH_OPS = {
            "sum": dist.ReduceOp.SUM,
            "min": dist.ReduceOp.MIN,
            "max": dist.ReduceOp.MAX,
            "prod": dist.ReduceOp.PRODUCT,
        }
    return _TORCH_OPS[op]


def all_reduce(tensor, op="sum", name=None, average=False, async_op=False):
    """All-reduce a tensor in place; returns the tensor (or a work handle
    tuple when async_op on the RCCL path)."""
    _ensure_init()
    if _core.size() == 1:
        return tensor
    if tensor.is_cuda:
        nat = _native()
        if nat is not None:
            h = nat.all_reduce_async(tensor, op=op, name=name or "")
            if async_op:
                if average:
                    raise ValueError("average not supported with async_op")
                return tensor, _NativeWork(h)
            nat.wait(h)
            if average:
                tensor.div_(_core.size())
            return tensor
        dist = _dist()
        work = dist.all_reduce(tensor, op=_torch_reduce_op(op),
                               async_op=async_op)
        if average:
            if async_op:
                raise ValueError("average not supported with async_op")
            tensor.div_(_core.size())
        return (tensor, work) if async_op else tensor
    t = tensor.contiguous()
    _core.all_reduce(t.data_ptr(), t.data_ptr(), t.numel(),
                     core_dtype(t.dtype), core_op(op),
                     name or _auto_name("ar", t.numel()))
    if t.data_ptr() != tensor.data_ptr():
        tensor.copy_(t)
    if average:
        tensor.div_(_core.size())
    return tensor


def broadcast(tensor, root=0, name=None):
    _ensure_init()
    if _core.size() == 1:
        return tensor
    if tensor.is_cuda:
        nat = _native()
        if nat is not None:
            return nat.broadcast(tensor, root=root, name=name or "")
        _dist().broadcast(tensor, src=root)
        return tensor
    t = tensor.contiguous()
    _core.broadcast(t.data_ptr(), t.data_ptr(), t.numel(),
                    core_dtype(t.dtype),
                    name or _auto_name("bc", t.numel()), root)
    if t.data_ptr() != tensor.data_ptr():
        tensor.copy_(t)
    return tensor


def all_gather(tensor, name=None):
    """Returns a new tensor of shape (size,) + tensor.shape."""
    _ensure_init()
    np_ = _core.size()
    out = torch.empty((np_,) + tuple(tensor.shape), dtype=tensor.dtype,
                      device=tensor.device)
    if np_ == 1:
        out[0] = tensor
        return out
    if tensor.is_cuda:
        nat = _native()
        if nat is not None:
            nat.wait(nat.all_gather_async(out.view(np_, -1),
                                          tensor.contiguous().view(-1),
                                          name=name or ""))
            return out
        dist = _dist()
        dist.all_gather_into_tensor(out.view(np_, -1),
                                    tensor.contiguous().view(1, -1))
        return out
    t = tensor.contiguous()
    _core.all_gather(t.data_ptr(), out.data_ptr(), t.numel(),
                     core_dtype(t.dtype),
                     name or _auto_name("ag", t.numel()))
    return out


def reduce(tensor, op="sum", name=None):
    """Reduce to rank 0 (in place at root; other ranks' buffers are used as
    scratch)."""
    _ensure_init()
    if _core.size() == 1:
        return tensor
    if tensor.is_cuda:
        nat = _native()
        if nat is not None:
            return nat.reduce(tensor, op=op, root=0, name=name or "")
        _dist().reduce(tensor, dst=0, op=_torch_reduce_op(op))
        return tensor
    t = tensor.contiguous()
    _core.reduce(t.data_ptr(), t.data_ptr(), t.numel(),
                 core_dtype(t.dtype), core_op(op),
                 name or _auto_name("rd", t.numel()))
    if t.data_ptr() != tensor.data_ptr():
        tensor.copy_(t)
    return tensor


# ---- CPU-staged variants (GPU tensor through host into the C++ engine) ----

def cpu_staged_all_reduce(tensor, op="sum", name=None):
    _ensure_init()
    if _core.size() == 1:
        return tensor
    host = tensor.detach().to("cpu", non_blocking=False).contiguous()
    _core.all_reduce(host.data_ptr(), host.data_ptr(), host.numel(),
                     core_dtype(host.dtype), core_op(op),
                     name or _auto_name("h", host.numel()))
    tensor.copy_(host.to(tensor.device))
    return tensor


_hier_groups = {}


def _hierarchical_groups():
    """Build (once per cluster version) the torch.distributed sub-groups
    for hierarchical all-reduce: one group per host, plus one group of the
    local masters. Every rank must enter every new_group call
    (torch.distributed contract), so group construction is itself a
    collective."""
    import torch.distributed as dist

    ver = int(_core.cluster_version())
    cached = _hier_groups.get("v")
    if cached is not None and cached[0] == ver:
        return cached[1]
    import json

    # host layout from the control plane: allgather each rank's host ip
    world = _core.size()
    my_host = _core.uid().rsplit(":", 1)[0]
    import numpy as np

    enc = np.zeros(16, dtype=np.uint8)
    raw = my_host.encode()[:16]
    enc[:len(raw)] = np.frombuffer(raw, dtype=np.uint8)
    out = np.zeros(16 * world, dtype=np.uint8)
    _core.all_gather(enc.ctypes.data, out.ctypes.data, 16, 0, "|hiermap")
    hosts = [bytes(out[i * 16:(i + 1) * 16]).rstrip(b"\0").decode()
             for i in range(world)]
    order = []
    for h in hosts:
        if h not in order:
            order.append(h)
    local_groups = {}
    masters = []
    for h in order:
        ranks = [r for r in range(world) if hosts[r] == h]
        masters.append(ranks[0])
        local_groups[h] = dist.new_group(ranks=ranks)
    cross_group = dist.new_group(ranks=masters)
    info = {
        "local": local_groups[my_host],
        "local_master": min(r for r in range(world)
                            if hosts[r] == my_host),
        "is_master": _core.rank() in masters,
        "cross": cross_group,
        "json": json.dumps({"hosts": hosts, "masters": masters}),
    }
    _hier_groups["v"] = (ver, info)
    return info


def hierarchical_all_reduce(tensor, name=None):
    """Local (intra-host) reduce to the local master -> cross-host
    all-reduce among masters -> local broadcast (reference
    ops/gpu/collective.cpp:106-158 ScheduledHierarchicalNcclAllReduce).
    On one host this degrades to a plain all-reduce. With a live
    torch.distributed process group the three hops run on sub-groups
    (RCCL on GPUs); otherwise the C++ engine's local/cross/local
    strategies carry it."""
    import torch.distributed as dist

    _ensure_init()
    if _core.host_count() <= 1:
        return all_reduce(tensor, name=name)
    if tensor.is_cuda:
        nat = _native()
        if nat is not None:
            # native sub-communicators (LOCAL + CROSS scopes) — no torch
            # sub-groups involved
            return nat.hierarchical_all_reduce(tensor, name=name or "")
    if dist.is_available() and dist.is_initialized():
        g = _hierarchical_groups()
        dist.reduce(tensor, dst=g["local_master"], group=g["local"])
        if g["is_master"]:
            dist.all_reduce(tensor, group=g["cross"])
        dist.broadcast(tensor, src=g["local_master"], group=g["local"])
        return tensor
    if tensor.is_cuda:
        # no process group: stage through the host into the C++ engine
        # (the reference's CrossAllReduceGpu D2H path)
        return cpu_staged_all_reduce(tensor, name=name)
    t = tensor.contiguous()
    _core.local_reduce(t.data_ptr(), t.data_ptr(), t.numel(),
                       core_dtype(t.dtype), core_op("sum"),
                       (name or "t") + "|lr")
    _core.cross_all_reduce(t.data_ptr(), t.data_ptr(), t.numel(),
                           core_dtype(t.dtype), core_op("sum"),
                           (name or "t") + "|x")
    _core.local_broadcast(t.data_ptr(), t.data_ptr(), t.numel(),
                          core_dtype(t.dtype), (name or "t") + "|lb")
    if t.data_ptr() != tensor.data_ptr():
        tensor.copy_(t)
    return tensor


def monitored_all_reduce(tensor, tree=None, op="sum", name=None):
    """All-reduce with per-strategy throughput monitoring and an optional
    per-call spanning tree (reference ops/collective.py:27-44
    monitored_all_reduce / session AllReduceWith; passing a tree per call
    is deprecated upstream in favor of set_tree, both are supported).
    CPU engine only — on CUDA tensors without a tree this is the plain
    (monitored) all_reduce."""
    _ensure_init()
    if _core.size() == 1:
        return tensor
    if tree is None:
        return all_reduce(tensor, op=op, name=name)
    if tensor.is_cuda:
        raise ValueError("per-call trees run on the CPU engine; use "
                         "set_tree for the GPU path")
    t = tensor.contiguous()
    _core.all_reduce_with([int(p) for p in tree], t.data_ptr(),
                          t.data_ptr(), t.numel(), core_dtype(t.dtype),
                          core_op(op), name or _auto_name("mar", t.numel()))
    if t.data_ptr() != tensor.data_ptr():
        tensor.copy_(t)
    return tensor


# ---- model-level helpers ----

def broadcast_parameters(params, root=0):
    """Broadcast model parameters (and optionally buffers) from root.
    Reference: kungfu/tensorflow/initializer BroadcastGlobalVariablesOp."""
    _ensure_init()
    if _core.size() == 1:
        return
    params = list(params)
    if not params:
        return
    if params[0].is_cuda:
        nat = _native()
        if nat is not None:
            handles = [nat.broadcast_async(p.data, root=root,
                                           name="bcast/%d" % i)
                       for i, p in enumerate(params)]
            for h in handles:
                nat.wait(h)
            return
        dist = _dist()
        for p in params:
            dist.broadcast(p.data, src=root)
        return
    for i, p in enumerate(params):
        broadcast(p.data, root=root, name="bcast/%d" % i)


def broadcast_model(model, root=0):
    with torch.no_grad():
        broadcast_parameters(list(model.parameters()), root=root)
        for b in model.buffers():
            broadcast(b.data, root=root)


# ---- P2P model store (AD-PSGD) ----

def save_tensor(name, tensor):
    _ensure_init()
    t = tensor.detach()
    if t.is_cuda:
        t = t.to("cpu")
    t = t.contiguous()
    _core.save(name, t.data_ptr(), t.numel() * t.element_size())


def request_tensor(target_rank, name, out):
    """Pull a stored tensor from target_rank into `out`; returns success."""
    _ensure_init()
    if out.is_cuda:
        host = torch.empty(out.shape, dtype=out.dtype, device="cpu")
        ok = _core.request(int(target_rank), name, host.data_ptr(),
                           host.numel() * host.element_size())
        if ok:
            out.copy_(host.to(out.device))
        return ok
    o = out.contiguous()
    ok = _core.request(int(target_rank), name, o.data_ptr(),
                       o.numel() * o.element_size())
    if ok and o.data_ptr() != out.data_ptr():
        out.copy_(o)
    return ok


_egress_last = {}


def egress_rates(window_hint=None):
    """Per-peer egress rates in bytes/sec since the previous call
    (reference: EgressRates op, ops/cpu/monitoring.cpp + monitor rates)."""
    import time

    global _egress_last
    _ensure_init()
    now = time.time()
    cur = _core.egress_bytes()
    rates = {}
    prev, t0 = _egress_last.get("snap", ({}, now))
    dt = max(now - t0, 1e-6)
    for peer_str, total in cur.items():
        rates[peer_str] = (total - prev.get(peer_str, 0)) / dt
    _egress_last["snap"] = (cur, now)
    return rates


# ---- adaptive topology ----

def get_peer_latencies():
    _ensure_init()
    return _core.peer_latencies_us()


def minimum_spanning_tree(latency_matrix_flat, n):
    return _core.prim_mst([float(x) for x in latency_matrix_flat], int(n))


def set_tree(parent):
    _ensure_init()
    _core.set_tree([int(p) for p in parent])


def compute_mst_tree():
    """Measure pairwise latencies (allgathered) and install the MST as the
    collective topology (reference: GetPeerLatencies + MinimumSpanningTree +
    SetTree, ops/adapt.py:49-60)."""
    import numpy as np

    _ensure_init()
    np_ = _core.size()
    lat = np.asarray(_core.peer_latencies_us(), dtype=np.float64)
    lat = np.maximum(lat, 1.0)
    row = torch.from_numpy(lat)
    mat = all_gather(row, name="|latmat").numpy()
    mat = np.maximum(mat, mat.T)  # symmetrize
    parent = minimum_spanning_tree(mat.flatten().tolist(), np_)
    set_tree(parent)
    return parent
