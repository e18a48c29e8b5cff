"""Pair averaging (AD-PSGD): asynchronous gossip model exchange.

Reference parity: srcs/python/kungfu/tensorflow/optimizers/async_sgd.py +
the P2P store/request ops (ops/cpu/p2p_new.cpp, peer_to_peer.cpp). Each
step: pull a random peer's stored model, v <- 0.5*(v + other), apply local
gradients, publish the updated model to the local blob store for other
peers to pull. No global synchronization — peers progress independently.

MI355X-native notes: the model is fused into one flat tensor by the HIP
pack kernel; the pull is served by the C++ control plane's request/response
endpoint (colocated peers ride Unix sockets); an optional prefetch thread
overlaps the pull with compute (reference's AsyncRequestModel
double-buffer, peer_to_peer.cpp:411-520).
"""
import random
import threading

import torch

from kungfu_amd import _core
from kungfu_amd.optimizers.core import KungFuOptimizer
from kungfu_amd.parallel.fusion import FlatParamGroup

_MODEL_KEY = "model"


def tournament_partner(rank, step, n):
    """Symmetric pairing for round `step` (circle-method round-robin):
    every rank's partner has this rank as its own partner, and over n-1
    (n even) rounds each rank meets every other exactly once. Returns -1
    when this rank sits out (odd n). Used by the RCCL/xGMI exchange mode
    where both sides of a pair must participate in send/recv."""
    if n < 2:
        return -1
    q = n if n % 2 else n - 1  # odd modulus; with even n, rank n-1 floats
    m = step % q
    if n % 2 == 0 and rank == n - 1:
        return (m * (q + 1) // 2) % q
    cand = (m - rank) % q
    if cand == rank:
        return n - 1 if n % 2 == 0 else -1  # paired with the floater
    return cand


class PairAveragingOptimizer(KungFuOptimizer):
    def __init__(self, optimizer, fuse_requests=True, prefetch=False,
                 name=_MODEL_KEY, peer_selection="random",
                 exchange="store"):
        """fuse_requests: accepted for reference-API compatibility; the
        model is always exchanged as one fused flat buffer here.
        peer_selection: 'random' (AD-PSGD gossip) or 'roundrobin'
        (reference GetNeighbour/RoundRobin ops, ops/cpu/topology.cpp).
        exchange: 'store' (asymmetric pulls from the P2P blob store, the
        reference's model) or 'rccl' (symmetric tournament gossip with
        direct GPU-GPU sendrecv over xGMI — both partners exchange and
        average on-device; needs the torch.distributed process group)."""
        super().__init__(optimizer)
        self.name = name
        self.prefetch = prefetch
        self.peer_selection = peer_selection
        self.exchange = exchange
        self._recv_buf = None
        self._rr_step = 0
        self._group = FlatParamGroup(self._params())
        self._host_buf = torch.empty(self._group.numel,
                                     dtype=self._group.dtype, device="cpu")
        self._other_host = torch.empty_like(self._host_buf)
        self._init_done = False
        self._prefetch_thread = None
        self._prefetch_ok = False

    def _publish(self):
        g = self._group
        g.pack()
        src = g.flat
        if src.is_cuda:
            self._host_buf.copy_(src.to("cpu"))
            host = self._host_buf
        else:
            host = src.contiguous()
        _core.save(self.name, host.data_ptr(),
                   host.numel() * host.element_size())

    def _pick_peer(self):
        np_, r = _core.size(), _core.rank()
        if np_ <= 1:
            return -1
        if self.peer_selection == "roundrobin":
            self._rr_step += 1
            off = 1 + (self._rr_step - 1) % (np_ - 1)
            return (r + off) % np_
        t = random.randrange(np_ - 1)
        return t if t < r else t + 1

    def _pull(self, target):
        buf = self._other_host
        return _core.request(target, self.name, buf.data_ptr(),
                             buf.numel() * buf.element_size())

    def _start_prefetch(self):
        target = self._pick_peer()
        if target < 0:
            return

        def run():
            self._prefetch_ok = self._pull(target)

        self._prefetch_thread = threading.Thread(target=run, daemon=True)
        self._prefetch_thread.start()

    @torch.no_grad()
    def _rccl_exchange(self):
        """xGMI-native gossip: symmetric tournament pairing, direct
        GPU-to-GPU model sendrecv over RCCL (no host staging), fused
        on-device averaging. Both partners run the same schedule, so the
        exchange is deadlock-free by construction."""
        n, r = _core.size(), _core.rank()
        partner = tournament_partner(r, self._rr_step, n)
        self._rr_step += 1
        if partner < 0:
            self.optimizer.step()
            return
        g = self._group
        g.pack()  # flat <- v
        if self._recv_buf is None:
            self._recv_buf = torch.empty_like(g.flat)
        from kungfu_amd.ops import _native

        nat = _native()
        if nat is not None:
            nat.wait(nat.send_recv_async(g.flat, self._recv_buf, partner,
                                         name="gossip"))
        else:
            import torch.distributed as dist

            ops = [dist.P2POp(dist.isend, g.flat, partner),
                   dist.P2POp(dist.irecv, self._recv_buf, partner)]
            for work in dist.batch_isend_irecv(ops):
                work.wait()
        if g.flat.is_cuda:
            from kungfu_amd.ops import hip as hip_ops

            hip_ops.avg_inplace(g.flat, self._recv_buf, alpha=0.5)
        else:
            g.flat.mul_(0.5).add_(self._recv_buf, alpha=0.5)
        g.unpack()
        self.optimizer.step()

    @torch.no_grad()
    def _step(self):
        if _core.size() == 1:
            self.optimizer.step()
            return
        if self.exchange == "rccl":
            self._rccl_exchange()
            return
        if not self._init_done:
            # step 0: publish the initial model, then a barrier so every
            # peer has something to serve (reference async_sgd.py:106-108)
            self._publish()
            _core.barrier()
            self._init_done = True
            if self.prefetch:
                self._start_prefetch()

        g = self._group
        got = False
        if self.prefetch and self._prefetch_thread is not None:
            self._prefetch_thread.join()
            got = self._prefetch_ok
            self._prefetch_thread = None
        else:
            target = self._pick_peer()
            got = target >= 0 and self._pull(target)

        if got:
            g.pack()  # flat <- v
            other = self._other_host
            if g.flat.is_cuda:
                other = other.to(g.flat.device, non_blocking=False)
            if g.flat.is_cuda:
                from kungfu_amd.ops import hip as hip_ops

                hip_ops.avg_inplace(g.flat, other, alpha=0.5)
            else:
                g.flat.mul_(0.5).add_(other, alpha=0.5)
            g.unpack()

        self.optimizer.step()
        self._publish()
        if self.prefetch:
            self._start_prefetch()
