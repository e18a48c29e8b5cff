"""Adaptive communication demo: monitored all-reduce throughput drives an
interference vote that switches the collective topology cluster-wide;
optionally installs a latency-MST tree.

Reference parity: the adaptation experiments
(session/adaptiveStrategies.go, ops/adapt.py set_tree/monitored flow).

Run:  python -m kungfu_amd.run -np 4 -strategy STAR \
          python examples/adaptive_trainer.py --steps 12
"""
import argparse

import torch

import kungfu_amd as kf
from kungfu_amd.models import SLP
from kungfu_amd.ops import all_reduce, broadcast_model, compute_mst_tree
from kungfu_amd.optimizers import SynchronousSGDOptimizer
from kungfu_amd.parallel.adaptive import (check_interference_and_switch,
                                          print_strategy_stats)


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--steps", type=int, default=12)
    p.add_argument("--check-every", type=int, default=4)
    p.add_argument("--interference-ratio", type=float, default=0.8)
    p.add_argument("--mst", action="store_true",
                   help="install the latency MST as the topology first")
    args = p.parse_args()

    kf.init(with_torch=False)
    if args.mst and kf.size() > 1:
        parent = compute_mst_tree()
        if kf.rank() == 0:
            print("MST parent array:", parent, flush=True)
    torch.manual_seed(0)
    model = SLP(in_features=64, classes=8)
    broadcast_model(model)
    opt = SynchronousSGDOptimizer(
        torch.optim.SGD(model.parameters(), lr=0.05))

    for step in range(args.steps):
        x = torch.randn(16, 1, 8, 8)
        y = torch.randint(0, 8, (16,))
        opt.zero_grad()
        torch.nn.functional.cross_entropy(model(x), y).backward()
        opt.step()
        # extra monitored traffic so the stats window has signal
        probe = torch.ones(200_000)
        all_reduce(probe, name="probe")
        if (step + 1) % args.check_every == 0:
            switched = check_interference_and_switch(
                ratio=args.interference_ratio)
            if kf.rank() == 0:
                print_strategy_stats()
                if switched:
                    print("SWITCHED to %s at step %d" % (switched, step),
                          flush=True)
    print("ADAPT-DONE rank=%d" % kf.rank(), flush=True)
    kf.finalize()


if __name__ == "__main__":
    main()
