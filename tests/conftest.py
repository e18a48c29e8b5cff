import os
import sys

import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import random as _random

# Listen ports must stay BELOW the kernel's ephemeral client-port range
# (32768-60999 here): an in-flight outgoing socket can otherwise squat a
# randomly-chosen listen port and fail a worker's bind.
_PORT_COUNTER = [_random.randrange(20000, 30000, 64)]


def _ancestors():
    """PIDs in our own process chain (never reap those)."""
    chain = set()
    pid = os.getpid()
    while pid > 1:
        chain.add(pid)
        try:
            with open("/proc/%d/stat" % pid) as f:
                pid = int(f.read().split(")")[-1].split()[1])
        except (OSError, ValueError, IndexError):
            break
    return chain


def _reap_stale_workers():
    """Session-start hygiene: a previously interrupted test run can leave
    worker processes holding control-plane ports, which poisons later
    randomly-chosen port blocks. Reap only processes that are (a) our own
    repo's test workers (KUNGFU_SELF_SPEC in env AND this repo in the
    command line) and (b) not in this process's ancestor chain."""
    import glob
    import signal

    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    keep = _ancestors()
    for pid_dir in glob.glob("/proc/[0-9]*"):
        try:
            pid = int(pid_dir.rsplit("/", 1)[1])
        except ValueError:
            continue
        if pid in keep:
            continue
        try:
            with open(pid_dir + "/cmdline") as f:
                cmd = f.read().replace("\0", " ")
            if root not in cmd and "kungfu_amd" not in cmd:
                continue
            with open(pid_dir + "/environ", "rb") as f:
                env = f.read()
            if b"KUNGFU_SELF_SPEC=" not in env:
                continue
            os.kill(pid, signal.SIGKILL)
        except (OSError, PermissionError):
            continue


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires an MI355X GPU (run via gpurun)")
    _reap_stale_workers()


@pytest.fixture
def port_block():
    """A fresh block of 64 loopback ports per test."""
    base = _PORT_COUNTER[0]
    _PORT_COUNTER[0] += 64
    return base
