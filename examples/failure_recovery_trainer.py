"""Failure-recovery demo: heartbeat-monitored training that crashes once
and is restarted by `kungfu-run -auto-recover` with adjusted epochs.

Reference parity: examples/Failure_recovery_examples/* + the monitor
protocol (docs/monitor_proposal.md): the worker wraps every batch in
monitor_batch_begin/end, reports epochs, honors --restart by reloading its
checkpoint, and sends trainend when finished.
"""
import argparse
import os

import torch

import kungfu_amd as kf
from kungfu_amd.cmd import (monitor_batch_begin, monitor_batch_end,
                            monitor_epoch_end, monitor_train_end)
from kungfu_amd.models import SLP
from kungfu_amd.optimizers import SynchronousSGDOptimizer


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--n-epochs", type=int, default=4)
    p.add_argument("--restart", type=int, default=0)
    p.add_argument("--crash-at-epoch", type=int, default=-1)
    p.add_argument("--ckpt", default="/tmp/kungfu_fr_ckpt.pt")
    p.add_argument("--device", default="cpu")
    args = p.parse_args()

    kf.init(with_torch=False)
    torch.manual_seed(0)
    model = SLP(in_features=16, classes=4).to(args.device)
    start_epoch = 0
    if args.restart and os.path.exists(args.ckpt + ".%d" % kf.rank()):
        state = torch.load(args.ckpt + ".%d" % kf.rank(),
                           map_location=args.device)
        model.load_state_dict(state["model"])
        start_epoch = state["epoch"]
        print("RESTARTED from epoch %d" % start_epoch, flush=True)
    opt = SynchronousSGDOptimizer(
        torch.optim.SGD(model.parameters(), lr=0.05))

    for epoch in range(args.n_epochs):
        real_epoch = start_epoch + epoch
        for _ in range(4):
            monitor_batch_begin()
            x = torch.randn(8, 1, 4, 4, device=args.device)
            y = torch.randint(0, 4, (8,), device=args.device)
            opt.zero_grad()
            torch.nn.functional.cross_entropy(model(x), y).backward()
            opt.step()
            monitor_batch_end()
        torch.save({"model": model.state_dict(), "epoch": real_epoch + 1},
                   args.ckpt + ".%d" % kf.rank())
        monitor_epoch_end()
        print("EPOCH %d done rank=%d" % (real_epoch, kf.rank()),
              flush=True)
        if (args.crash_at_epoch >= 0 and not args.restart
                and real_epoch + 1 == args.crash_at_epoch
                and kf.rank() == 0):
            print("CRASHING rank 0 now", flush=True)
            os._exit(1)  # simulated failure: process dies without cleanup
    monitor_train_end()
    print("TRAIN END rank=%d total_epochs=%d" %
          (kf.rank(), start_epoch + args.n_epochs), flush=True)
    kf.finalize()


if __name__ == "__main__":
    main()
