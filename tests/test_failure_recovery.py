"""Failure detection + auto-recovery (the fork's marquee feature):
kungfu-run -auto-recover restarts crashed training with adjusted epochs
(reference: runner/monitored.go + monitorserver/monitor.go)."""
import os
import re
import subprocess
import sys
from mp_helpers import retry_flaky

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


@retry_flaky
def test_auto_recover_restarts_after_crash(port_block, tmp_path):
    env = dict(os.environ)
    env["PYTHONPATH"] = ROOT + os.pathsep + env.get("PYTHONPATH", "")
    ckpt = str(tmp_path / "ckpt.pt")
    import collections

    from mp_helpers import run_launcher_graceful

    rc, stdout, stderr = run_launcher_graceful(
        [sys.executable, "-m", "kungfu_amd.run",
         "-np", "2", "-port", str(port_block), "-port-range",
         str(port_block + 1), "-auto-recover", "3s",
         "-monitor-port", str(port_block + 60),
         sys.executable, "examples/failure_recovery_trainer.py",
         "--n-epochs", "4", "--crash-at-epoch", "2", "--ckpt", ckpt],
        ROOT, env, 300)
    r = collections.namedtuple("R", "returncode stdout stderr")(
        rc, stdout, stderr)
    out = re.sub(r"\x1b\[[0-9;]*m", "", r.stdout)
    assert "CRASHING rank 0 now" in out
    assert "failure detected" in out
    assert out.count("RESTARTED from epoch 2") == 2
    ends = [ln for ln in out.splitlines() if "TRAIN END" in ln]
    assert len(ends) == 2, out
    assert all("total_epochs=4" in ln for ln in ends)
    assert r.returncode == 0, out + r.stderr
