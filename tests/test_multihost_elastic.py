"""Two-runner multi-host elastic e2e on loopback aliases: one kungfu-run
per 'host' (127.0.0.1 / 127.0.0.2), shared standalone config server,
schedule-driven resize that grows workers on BOTH hosts — validates the
multi-runner stage-notification path (reference: cluster-in-docker tests +
watch.go across hosts)."""
import os
import re
import signal
import subprocess
import sys
import time

from mp_helpers import run_launcher_graceful  # noqa: F401
from mp_helpers import retry_flaky

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _communicate_all(procs, deadline):
    """communicate() with graceful SIGTERM on timeout so runners clean up
    their worker process groups (no orphan port squatters)."""
    outs = []
    for p in procs:
        try:
            out, err = p.communicate(timeout=max(5, deadline - time.time()))
            outs.append((p.returncode, _plain(out), _plain(err)))
        except subprocess.TimeoutExpired:
            p.send_signal(signal.SIGTERM)
            try:
                out, err = p.communicate(timeout=40)
            except subprocess.TimeoutExpired:
                p.kill()
                out, err = p.communicate()
            outs.append((124, _plain(out or ""), _plain(err or "")))
    return outs


def _plain(s):
    return re.sub(r"\x1b\[[0-9;]*m", "", s)


@retry_flaky
def test_two_runners_elastic_grow(port_block):
    from kungfu_amd.launcher.configserver import make_server

    cfg_port = port_block + 60
    hosts = "127.0.0.1:3,127.0.0.2:3"
    env = dict(os.environ)
    env["PYTHONPATH"] = ROOT + os.pathsep + env.get("PYTHONPATH", "")

    import json

    from kungfu_amd import _core

    peers = _core.gen_peer_list(hosts, 4, port_block + 2)
    runners = _core.gen_runner_list(hosts, port_block)
    srv, thread = make_server(cfg_port, json.dumps(
        {"runners": runners.split(","), "workers": peers.split(",")}))
    thread.start()
    try:
        procs = []
        for self_ip in ("127.0.0.1", "127.0.0.2"):
            procs.append(subprocess.Popen(
                [sys.executable, "-m", "kungfu_amd.run",
                 "-np", "4", "-H", hosts, "-self", self_ip,
                 "-port", str(port_block), "-port-range",
                 str(port_block + 2),
                 "-config-server", "127.0.0.1:%d" % cfg_port, "-w",
                 sys.executable, "examples/elastic_trainer.py",
                 "--schedule", "3:6", "--max-step", "7"],
                cwd=ROOT, env=env, stdout=subprocess.PIPE,
                stderr=subprocess.PIPE, text=True))
        outs = _communicate_all(procs, time.time() + 240)
        for rc, out, err in outs:
            assert rc == 0, out + err
        all_out = "\n".join(o for _, o, _ in outs)
        done = [ln for ln in all_out.splitlines() if "DONE" in ln]
        assert len(done) == 6, all_out  # 4 initial + 2 joiners finish
        assert all("size=6 step=7" in ln for ln in done)
        # joiners were placed by least-loaded host -> host 2's runner
        # spawned them (it started with 1 worker)
        assert all_out.count("JOIN") == 6
        resized = [ln for ln in all_out.splitlines() if "RESIZED" in ln]
        assert len(resized) == 4  # the four survivors each re-synced
    finally:
        srv.shutdown()


@retry_flaky
def test_two_runners_auto_recover(port_block):
    """Cross-host failure recovery: a worker on host 2 crashes; host 1's
    monitor detects the stall (all heartbeats go to host 0 = runners[0]),
    broadcasts otherdown to host 2's monitor, and BOTH runners restart
    their workers with adjusted epochs (reference monitored.go +
    monitor.go otherdown flow)."""
    hosts = "127.0.0.1:2,127.0.0.2:2"
    env = dict(os.environ)
    env["PYTHONPATH"] = ROOT + os.pathsep + env.get("PYTHONPATH", "")
    ckpt = "/tmp/kf_mh_fr_%d.pt" % port_block
    procs = []
    for self_ip in ("127.0.0.1", "127.0.0.2"):
        procs.append(subprocess.Popen(
            [sys.executable, "-m", "kungfu_amd.run",
             "-np", "4", "-H", hosts, "-self", self_ip,
             "-port", str(port_block), "-port-range", str(port_block + 2),
             "-auto-recover", "3s", "-monitor-port", str(port_block + 58),
             sys.executable, "examples/failure_recovery_trainer.py",
             "--n-epochs", "4", "--crash-at-epoch", "2", "--ckpt", ckpt],
            cwd=ROOT, env=env, stdout=subprocess.PIPE,
            stderr=subprocess.PIPE, text=True))
    outs = _communicate_all(procs, time.time() + 240)
    all_out = "\n".join(o for _, o, _ in outs)
    for rc, out, err in outs:
        assert rc == 0, all_out + err
    assert "CRASHING rank 0 now" in all_out
    # both runners restarted their workers
    assert all_out.count("RESTARTED from epoch 2") == 4, all_out
    ends = [ln for ln in all_out.splitlines() if "TRAIN END" in ln]
    assert len(ends) == 4 and all("total_epochs=4" in ln for ln in ends)
