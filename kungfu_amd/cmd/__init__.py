"""Worker-side monitor signals + embedded launcher.

Reference parity: srcs/python/kungfu/cmd/__init__.py — monitor_batch_begin/
end, monitor_epoch_end, monitor_train_end POST heartbeats to the first
host's monitor server on :7756 (libkungfu-comm/send.go:20-63), and run()
embeds the launcher.
"""
import json
import os
import urllib.request

from kungfu_amd.launcher.monitor import MONITOR_PORT


def _monitor_port():
    return int(os.environ.get("KUNGFU_MONITOR_PORT", MONITOR_PORT))


def _monitor_host():
    runners = os.environ.get("KUNGFU_INIT_RUNNERS", "")
    if runners:
        return runners.split(",")[0].rsplit(":", 1)[0]
    return "127.0.0.1"


def _rank():
    try:
        from kungfu_amd import rank

        return rank()
    except Exception:
        return 0


def _send(key):
    try:
        req = urllib.request.Request(
            "http://%s:%d/" % (_monitor_host(), _monitor_port()),
            data=json.dumps({"key": key}).encode(), method="POST")
        urllib.request.urlopen(req, timeout=2)
        return True
    except Exception:
        return False


def monitor_batch_begin():
    return _send("begin:%d" % _rank())


def monitor_batch_end():
    return _send("end:%d" % _rank())


def monitor_epoch_end():
    return _send("epoch:%d" % _rank())


def monitor_train_end():
    return _send("trainend:%d" % _rank())


def run(argv=None):
    """Embedded launcher (reference kungfu.cmd.run -> kungfu_run_main)."""
    from kungfu_amd.launcher.run import main

    main(argv)


def _mp_worker(fn, spec, peers, args):
    # module-level so the spawn context can pickle it (a local closure
    # cannot be); fn itself must be a module-level callable
    os.environ["KUNGFU_SELF_SPEC"] = spec
    os.environ["KUNGFU_INIT_PEERS"] = peers
    fn(*args)


def launch_multiprocess(fn, np, *args, port_base=34100):
    """Single-machine multiprocessing helper (reference
    launch_multiprocess): runs fn in np local worker processes via the
    launcher env protocol using multiprocessing (spawn)."""
    import multiprocessing as mp

    ctx = mp.get_context("spawn")
    peers = ",".join("127.0.0.1:%d" % (port_base + i) for i in range(np))
    procs = [
        ctx.Process(target=_mp_worker,
                    args=(fn, "127.0.0.1:%d" % (port_base + i), peers,
                          args))
        for i in range(np)
    ]
    for p in procs:
        p.start()
    for p in procs:
        p.join()
    return [p.exitcode for p in procs]
