"""kungfu-run launcher end-to-end on loopback (reference pattern:
run-integration-tests.sh / run-train-tests.sh under the real launcher)."""
import os
import subprocess
import sys
from mp_helpers import retry_flaky

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def run_launcher(args, timeout=180):
    import collections

    from mp_helpers import run_launcher_graceful

    env = dict(os.environ)
    env["PYTHONPATH"] = ROOT + os.pathsep + env.get("PYTHONPATH", "")
    rc, out, err = run_launcher_graceful(
        [sys.executable, "-m", "kungfu_amd.run"] + args, ROOT, env,
        timeout)
    R = collections.namedtuple("R", "returncode stdout stderr")
    return R(rc, out, err)


@retry_flaky
def test_kungfu_run_np2_mnist_slp(port_block):
    r = run_launcher([
        "-np", "2", "-port", str(port_block), "-port-range",
        str(port_block + 1), "-timeout", "150s",
        sys.executable, "examples/mnist_slp.py", "--n-epochs", "2",
    ])
    assert r.returncode == 0, r.stdout + r.stderr
    assert r.stdout.count("FINAL") == 2


@retry_flaky
def test_kungfu_run_env_protocol(port_block):
    script = ("import kungfu_amd as kf, os; kf.init(with_torch=False); "
              "print('R', kf.rank(), kf.size(), kf.local_rank(), "
              "os.environ.get('HIP_VISIBLE_DEVICES'))")
    r = run_launcher([
        "-np", "3", "-port", str(port_block), "-port-range",
        str(port_block + 1), "-strategy", "RING",
        sys.executable, "-c", script,
    ])
    assert r.returncode == 0, r.stdout + r.stderr
    import re

    plain = re.sub(r"\x1b\[[0-9;]*m", "", r.stdout)
    lines = sorted(ln.split("] ", 1)[1].strip()
                   for ln in plain.splitlines() if "] R " in ln)
    assert lines == ["R 0 3 0 0", "R 1 3 1 1", "R 2 3 2 2"], repr(plain)


@retry_flaky
def test_kungfu_run_propagates_failure(port_block):
    r = run_launcher([
        "-np", "2", "-port", str(port_block), "-port-range",
        str(port_block + 1),
        sys.executable, "-c", "import sys; sys.exit(3)",
    ])
    assert r.returncode == 3


def test_rrun_dry_run():
    """kungfu-rrun builds one ssh command per WORKER with the full env
    protocol (reference cmd/kungfu-rrun/rrun.go: workers launched
    directly, no per-host launcher)."""
    import subprocess
    import sys

    out = subprocess.run(
        [sys.executable, "-m", "kungfu_amd.launcher.rrun", "-np", "4",
         "-H", "10.0.0.1:2,10.0.0.2:2", "--dry-run", "--",
         "python3", "train.py"],
        capture_output=True, text=True, cwd=ROOT)
    assert out.returncode == 0, out.stderr
    lines = [ln for ln in out.stdout.splitlines() if ln.startswith("[")]
    assert len(lines) == 4
    assert "KUNGFU_SELF_SPEC=10.0.0.1:30100" in lines[0]
    assert "ssh" in lines[0] and "train.py" in lines[0]
    # round-robin-by-slot: ranks alternate hosts
    assert "10.0.0.2:30100" in lines[1]
    assert "KUNGFU_INIT_PEERS" in lines[0]
    assert "CUDA_VISIBLE_DEVICES=1" in lines[2]


def test_launch_multiprocess_helper(port_block, tmp_path):
    """kungfu_amd.cmd.launch_multiprocess (reference launch_multiprocess):
    runs fn in np local processes over the env protocol. (A file-based
    script: the spawn context must re-import __main__ to unpickle fn.)"""
    import subprocess
    import sys

    code = """
from kungfu_amd.cmd import launch_multiprocess

def work():
    import numpy as np
    import kungfu_amd as kf
    from kungfu_amd import _core
    kf.init(with_torch=False)
    a = np.ones(64, dtype=np.float32)
    _core.all_reduce(a.ctypes.data, a.ctypes.data, a.size, 10, 0, "lm")
    assert a[0] == 2.0, a[0]
    kf.finalize()

if __name__ == "__main__":
    codes = launch_multiprocess(work, 2, port_base=%d)
    assert codes == [0, 0], codes
    print("LMP OK")
""" % port_block
    script = tmp_path / "lmp.py"
    script.write_text(code)
    env = dict(os.environ)
    env["PYTHONPATH"] = ROOT
    out = subprocess.run([sys.executable, str(script)], env=env, cwd=ROOT,
                         capture_output=True, text=True, timeout=180)
    assert out.returncode == 0 and "LMP OK" in out.stdout, \
        out.stdout + out.stderr


@retry_flaky
def test_imagenet_resnet_example(port_block):
    """The flagship example (elastic trainer + heartbeats + checkpoint
    wiring + GNS probe) runs end-to-end on a tiny CPU config."""
    import re
    import subprocess
    import sys

    env = dict(os.environ)
    env["PYTHONPATH"] = ROOT + os.pathsep + env.get("PYTHONPATH", "")
    from mp_helpers import run_launcher_graceful

    rc, out, err = run_launcher_graceful(
        [sys.executable, "-m", "kungfu_amd.run", "-np", "2",
         "-port", str(port_block), "-port-range", str(port_block + 1),
         sys.executable, "examples/imagenet_resnet.py", "--epochs", "1",
         "--batch-size", "2", "--samples", "8", "--image-size", "64"],
        ROOT, env, 280)
    plain = re.sub(r"\x1b\[[0-9;]*m", "", out)
    assert rc == 0, plain + err
    assert plain.count("DONE rank=") == 2, plain


def test_hostfile_mpi_format(tmp_path):
    """-hostfile accepts the MPI-style format the reference parses
    (plan/hostfile/hostfile.go): 'ip slots=N [public_addr=X]' plus
    comments; bare ip:slots lines pass through."""
    from kungfu_amd.launcher.run import parse_hostfile

    text = ("# cluster\n"
            "10.0.0.1 slots=4\n"
            "10.0.0.2 slots=2 public_addr=1.2.3.4  # edge\n"
            "10.0.0.3\n"
            "10.0.0.4:8\n")
    assert parse_hostfile(text) == \
        "10.0.0.1:4,10.0.0.2:2:1.2.3.4,10.0.0.3:1,10.0.0.4:8"
    from kungfu_amd import _core

    peers = _core.gen_peer_list(parse_hostfile(text), 7, 30000)
    assert len(peers.split(",")) == 7


@retry_flaky
def test_allreduce_microbenchmark_cpu(port_block):
    """`python -m kungfu_amd.benchmarks --method CPU` under kungfu-run
    (reference `python -m kungfu.tensorflow.v1.benchmarks`): emits the
    algbw report line."""
    import re
    import sys

    env = dict(os.environ)
    env["PYTHONPATH"] = ROOT + os.pathsep + env.get("PYTHONPATH", "")
    from mp_helpers import run_launcher_graceful

    rc, out, err = run_launcher_graceful(
        [sys.executable, "-m", "kungfu_amd.run", "-np", "2",
         "-port", str(port_block), "-port-range", str(port_block + 1),
         sys.executable, "-m", "kungfu_amd.benchmarks", "--method", "CPU",
         "--model", "slp-mnist", "--steps", "5", "--warmup", "2"],
        ROOT, env, 200)
    plain = re.sub(r"\x1b\[[0-9;]*m", "", out)
    assert rc == 0, plain + err
    m = re.search(r"method=CPU .*np=2 .*algbw=([0-9.]+) GB/s", plain)
    assert m and float(m.group(1)) > 0, plain
