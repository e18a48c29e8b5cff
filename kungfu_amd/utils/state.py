"""Stateful training helpers.

Reference parity: the Counter and ExponentialMovingAverage TF ops
(src/tensorflow/ops/cpu/state.cpp) and the python timeline logger
(kungfu/_utils.py:44-50).
"""
import json
import os
import time


class StepCounter:
    def __init__(self, start=0, incr=1):
        self.value = int(start)
        self.incr = int(incr)

    def __call__(self):
        v = self.value
        self.value += self.incr
        return v


class ExponentialMovingAverage:
    def __init__(self, alpha=0.9):
        self.alpha = float(alpha)
        self.value = None

    def update(self, x):
        x = float(x)
        if self.value is None:
            self.value = x
        else:
            self.value = self.alpha * self.value + (1 - self.alpha) * x
        return self.value


def job_start_timestamp():
    return float(os.environ.get("KUNGFU_JOB_START_TIMESTAMP", time.time()))


def log_event(name, file=None):
    """Timeline event relative to job start (reference _utils.py)."""
    t = time.time() - job_start_timestamp()
    line = "[kungfu-event] %.6f %s" % (t, name)
    print(line, flush=True, file=file)
    return t


def dump_chrome_trace(path):
    """Write the C++ runtime trace (KUNGFU_ENABLE_TRACE=1) as a
    chrome://tracing JSON file."""
    from kungfu_amd import _core, rank

    events = [{
        "name": name,
        "ph": "X",
        "ts": start_us,
        "dur": dur_us,
        "pid": rank(),
        "tid": 0,
    } for name, start_us, dur_us in _core.trace_events()]
    with open(path, "w") as f:
        json.dump({"traceEvents": events}, f)
    return len(events)
