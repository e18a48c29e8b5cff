"""KungFu-AMD: MI355X-native adaptive distributed training runtime.

A from-scratch rebuild of the capabilities of DingtongHan/KungFu-1
(decentralized data-parallel training with elastic scaling, adaptive
communication topologies, monitoring, and failure recovery), designed
MI355X-first:

  * one process per GPU; GPU collectives run on RCCL over xGMI through
    ``torch.distributed`` (backend "nccl" is RCCL on ROCm), bucketed and
    overlapped with backward;
  * the control plane (consensus, elastic resize, P2P model store, CPU
    collectives over 8 graph topologies) is a C++17 runtime
    (``kungfu_amd._core``) replacing the reference's Go stack;
  * hot device ops (fusion pack/unpack, model averaging, norm², fused SGD)
    are hand-written gfx950 HIP kernels (``kungfu_amd._hip``).

Public surface mirrors the reference python API
(srcs/python/kungfu/python/__init__.py: rank/size/barrier/resize/...).
"""
import atexit
import os

from kungfu_amd import _core

__version__ = "0.1.0"

_initialized = False
_torch_dist = False


def _synthesize_env():
    """Allow running under plain ``torchrun`` (RANK/WORLD_SIZE/MASTER_*)
    without the kungfu-run launcher: build the control-plane peer list on
    127.0.0.1 with a deterministic port block."""
    if os.environ.get("KUNGFU_SELF_SPEC"):
        return
    world = int(os.environ.get("WORLD_SIZE", "1"))
    if world <= 1:
        return  # single-process fallback inside _core
    rank = int(os.environ["RANK"])
    base = int(os.environ.get("KUNGFU_PORT_BASE", "23100"))
    host = "127.0.0.1"
    peers = ",".join("%s:%d" % (host, base + i) for i in range(world))
    os.environ["KUNGFU_SELF_SPEC"] = "%s:%d" % (host, base + rank)
    os.environ["KUNGFU_INIT_PEERS"] = peers


def init(with_torch=True):
    """Initialize the runtime (idempotent).

    Starts the C++ control plane and, when CUDA(HIP) devices and a
    multi-process cluster are present, the native RCCL collective layer
    (communicators bootstrapped over the control plane itself — see
    kungfu_amd.ops.rccl). torch.distributed is only used as an explicit
    fallback (KUNGFU_GPU_BACKEND=torch) or for gloo-backed CPU tests
    (KUNGFU_TORCH_BACKEND=gloo).
    """
    global _initialized, _torch_dist
    if _initialized:
        return
    _synthesize_env()
    _core.init()
    _initialized = True
    if with_torch:
        _init_gpu_backend()


def _set_cuda_device():
    import torch

    rank, world = _core.rank(), _core.size()
    if torch.cuda.device_count() >= world:
        # launcher-less run (e.g. torchrun on one node)
        torch.cuda.set_device(int(os.environ.get("LOCAL_RANK", rank)))
    else:
        torch.cuda.set_device(0)  # kungfu-run pins HIP_VISIBLE_DEVICES


def _init_gpu_backend():
    """Bring up the GPU collective backend for the current cluster.

    Default: the native RCCL layer (kungfu_amd._rccl) with the uniqueId
    rendezvous over the C++ control plane — no TCPStore/env:// beside it.
    KUNGFU_GPU_BACKEND=torch selects the torch.distributed fallback;
    KUNGFU_TORCH_BACKEND (e.g. gloo) forces a torch process group for
    CPU-side tests.
    """
    if _core.size() <= 1:
        return
    import torch

    if os.environ.get("KUNGFU_TORCH_BACKEND"):  # tests: "gloo"
        _maybe_init_torch_dist()
        return
    if not torch.cuda.is_available():
        return  # CPU plumbing mode uses the C++ engine only
    if os.environ.get("KUNGFU_GPU_BACKEND", "rccl") == "torch":
        _maybe_init_torch_dist()
        return
    _set_cuda_device()
    from kungfu_amd.ops import rccl as rccl_ops

    try:
        rccl_ops.init_gpu()
    except Exception as e:
        # Insurance for unattended multi-GPU runs: a failed native
        # bootstrap (driver box quirk) must not kill the job silently —
        # log LOUDLY and fall back to the proven torch.distributed path.
        import sys

        print("[kungfu] native RCCL bootstrap FAILED (%s); falling back "
              "to torch.distributed" % e, file=sys.stderr, flush=True)
        try:
            rccl_ops.finalize()
        except Exception:
            pass
        _maybe_init_torch_dist()


def _maybe_init_torch_dist():
    """Bring up torch.distributed (fallback / gloo test backend)."""
    global _torch_dist
    if _torch_dist or _core.size() <= 1:
        return
    import torch

    backend = os.environ.get("KUNGFU_TORCH_BACKEND")  # tests: "gloo"
    if backend is None:
        if not torch.cuda.is_available():
            return  # CPU plumbing mode uses the C++ engine only
        backend = "nccl"  # RCCL on ROCm
    import torch.distributed as dist

    if dist.is_initialized():
        _torch_dist = True
        return
    rank, world = _core.rank(), _core.size()
    if torch.cuda.is_available():
        _set_cuda_device()
    if os.environ.get("MASTER_ADDR") and os.environ.get("MASTER_PORT"):
        # torchrun path: attach to the launcher's store (env://) — creating
        # our own TCPStore on MASTER_PORT would collide with torchrun's
        dist.init_process_group(backend=backend, rank=rank,
                                world_size=world)
    else:
        # kungfu-run path: rank 0 hosts a fresh store on a derived port
        # (re-derived per cluster version so elastic re-inits don't clash)
        first = os.environ["KUNGFU_INIT_PEERS"].split(",")[0]
        addr = first.rsplit(":", 1)[0]
        port = int(first.rsplit(":", 1)[1]) + 1711 + \
            int(_core.cluster_version())
        dist.init_process_group(
            backend=backend,
            init_method="tcp://%s:%d" % (addr, port),
            rank=rank,
            world_size=world,
        )
    _torch_dist = True


def _reinit_torch_dist():
    """Tear down and re-create the torch.distributed process group after a
    resize (reference: ResetNcclHelper, ops/gpu/scheduler.cpp:43-72).

    Gated on whether a group actually exists / could exist — NOT on CUDA
    availability: a gloo group (KUNGFU_TORCH_BACKEND=gloo) must also be
    destroyed and re-created, or survivors keep a stale world size while
    joiners block in init_process_group."""
    global _torch_dist
    import torch
    import torch.distributed as dist

    had_group = dist.is_available() and dist.is_initialized()
    if (not had_group and not torch.cuda.is_available()
            and not os.environ.get("KUNGFU_TORCH_BACKEND")):
        return  # pure CPU-plumbing mode: no process group in play
    if had_group:
        dist.destroy_process_group()
    _torch_dist = False
    if _core.size() > 1:
        _maybe_init_torch_dist()


def finalize():
    global _initialized, _torch_dist
    if not _initialized:
        return
    try:
        # the RCCL layer holds a capsule into _core: tear it down first
        from kungfu_amd.ops import rccl as rccl_ops

        rccl_ops.finalize()
    except Exception:
        pass
    try:
        import torch.distributed as dist

        if dist.is_available() and dist.is_initialized():
            dist.destroy_process_group()
    except Exception:
        pass
    _torch_dist = False
    _core.finalize()
    _initialized = False


atexit.register(finalize)


def _ensure_init():
    if not _initialized:
        init()


def rank():
    _ensure_init()
    return _core.rank()


def size():
    _ensure_init()
    return _core.size()


def local_rank():
    _ensure_init()
    return _core.local_rank()


def local_size():
    _ensure_init()
    return _core.local_size()


def host_count():
    _ensure_init()
    return _core.host_count()


def uid():
    _ensure_init()
    return _core.uid()


def detached():
    _ensure_init()
    return _core.detached()


def cluster_version():
    _ensure_init()
    return _core.cluster_version()


def run_barrier():
    """Control-plane barrier across all workers."""
    _ensure_init()
    _core.barrier()


barrier = run_barrier


def propose_new_size(new_size):
    _ensure_init()
    return _core.propose_new_size(int(new_size))


def resize(new_size=None):
    """Elastic resize. Returns (changed, detached).

    With an argument: rank 0 proposes `new_size` to the config server and
    all workers reconfigure. Without: re-read the config server state
    (resize_cluster_from_url semantics).
    """
    _ensure_init()
    if new_size is None:
        changed, det = _core.resize_cluster_from_url()
    else:
        changed, det = _core.resize(int(new_size))
    if changed and not det:
        from kungfu_amd.ops import rccl as rccl_ops

        if rccl_ops.active():
            # native path: rebuild communicators in place over the new
            # cluster (reference ResetNcclHelper) — no process-group
            # destroy/recreate
            rccl_ops.reinit()
        else:
            _reinit_torch_dist()
    return changed, det


def all_reduce_int_max(value):
    """Max-all-reduce a python int over the control plane (step sync)."""
    import numpy as np

    _ensure_init()
    buf = np.array([int(value)], dtype=np.int64)
    out = np.zeros(1, dtype=np.int64)
    _core.all_reduce(buf.ctypes.data, out.ctypes.data, 1, 4, 2, "|intmax")
    return int(out[0])


def consensus_bytes(data, name="consensus"):
    _ensure_init()
    return _core.consensus(bytes(data), name)
