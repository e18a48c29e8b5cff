"""Elastic training loop: resize the cluster mid-run on a step schedule.

Reference parity: the StepBasedSchedule op + KungFuElasticTrainHook flow
(ops/cpu/elastic.cpp:16-82, hooks/elastic.py): every step all workers call
resize(); rank 0 proposes the scheduled size; new workers join, sync the
step counter with a max-all-reduce, and receive the model by broadcast.
Run under `kungfu-run -w` (watch mode) with the builtin config server.
"""
import argparse

import torch

import kungfu_amd as kf
from kungfu_amd.models import SLP
from kungfu_amd.ops import broadcast_model
from kungfu_amd.optimizers import SynchronousSGDOptimizer


def parse_schedule(s):
    out = {}
    if s:
        for part in s.split(","):
            step, size = part.split(":")
            out[int(step)] = int(size)
    return out


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--schedule", default="", help="step:size,step:size")
    p.add_argument("--max-step", type=int, default=10)
    args = p.parse_args()
    schedule = parse_schedule(args.schedule)

    kf.init(with_torch=False)
    torch.manual_seed(0)
    model = SLP(in_features=16, classes=4)

    # joining workers learn the current step, then everyone takes the model
    # from rank 0 — the same [int_max, broadcast] pair the survivors run in
    # their post-resize branch, so the collective sequences line up
    step = kf.all_reduce_int_max(0)
    broadcast_model(model)
    opt = SynchronousSGDOptimizer(
        torch.optim.SGD(model.parameters(), lr=0.05))
    print("JOIN rank=%d size=%d step=%d version=%d" %
          (kf.rank(), kf.size(), step, kf.cluster_version()), flush=True)

    while step < args.max_step:
        x = torch.randn(8, 1, 4, 4)
        y = torch.randint(0, 4, (8,))
        opt.zero_grad()
        torch.nn.functional.cross_entropy(model(x), y).backward()
        opt.step()
        step += 1

        if kf.rank() == 0 and step in schedule:
            kf.propose_new_size(schedule[step])
        changed, detached = kf.resize()
        if detached:
            print("DETACHED rank_was=%s step=%d" % (kf.uid(), step),
                  flush=True)
            break
        if changed:
            # re-sync state across the new cluster (reference
            # hooks/elastic.py:49-58)
            step = kf.all_reduce_int_max(step)
            broadcast_model(model)
            opt = SynchronousSGDOptimizer(
                torch.optim.SGD(model.parameters(), lr=0.05))
            print("RESIZED size=%d step=%d version=%d" %
                  (kf.size(), step, kf.cluster_version()), flush=True)

    if not kf.detached():
        print("DONE rank=%d size=%d step=%d" % (kf.rank(), kf.size(), step),
              flush=True)
    kf.finalize()


def replace_cluster(new_workers):
    """Test helper: PUT an arbitrary worker list to the config server
    (exercises resizes the schedule API cannot express, e.g. removing
    rank 0)."""
    import json
    import os
    import urllib.request

    url = os.environ["KUNGFU_CONFIG_SERVER"]
    if not url.startswith("http"):
        url = "http://" + url
    body = json.dumps({
        "runners": os.environ.get("KUNGFU_INIT_RUNNERS", "").split(","),
        "workers": new_workers,
    }).encode()
    req = urllib.request.Request(url + "/config", data=body, method="PUT")
    urllib.request.urlopen(req, timeout=5)


if __name__ == "__main__":
    main()
