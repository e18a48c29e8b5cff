"""Fault injection for failure-recovery testing.

Reference parity: srcs/go/nccl/bug.go (RandomFailure simulating the NCCL
ld.so crash with probability p, wired to -rand-nccl-failure in
kungfu-bench-allreduce) and tests/go/cmd/kungfu-bad-worker.
"""
import os
import random
import sys


def random_failure(prob=None, exit_code=137, quiet=False):
    """Die with probability prob (default from KUNGFU_RAND_FAILURE)."""
    if prob is None:
        prob = float(os.environ.get("KUNGFU_RAND_FAILURE", "0"))
    if prob > 0 and random.random() < prob:
        if not quiet:
            print("[kungfu] injected random failure", flush=True)
        sys.stdout.flush()
        os._exit(exit_code)


def fail_at_step(step, fail_step, rank=None, only_rank=0, exit_code=1):
    """Deterministic crash at a given step (bad-worker pattern)."""
    if fail_step is None or fail_step < 0:
        return
    if step == fail_step and (rank is None or rank == only_rank):
        print("[kungfu] bad worker: crashing at step %d" % step,
              flush=True)
        os._exit(exit_code)
