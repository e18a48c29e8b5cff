"""Elastic resize end-to-end: kungfu-run watch mode + builtin config server
+ schedule-driven resize (reference: test_tensorflow_resize.py /
test_step_based_schedule.py under `kungfu-run -w`)."""
import os
import re
import subprocess
import sys

from mp_helpers import retry_flaky

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def run_watch(args, timeout=240):
    import collections

    from mp_helpers import run_launcher_graceful

    env = dict(os.environ)
    env["PYTHONPATH"] = ROOT + os.pathsep + env.get("PYTHONPATH", "")
    rc, out, err = run_launcher_graceful(
        [sys.executable, "-m", "kungfu_amd.run"] + args, ROOT, env,
        timeout)
    R = collections.namedtuple("R", "returncode stdout stderr")
    return R(rc, out, err)


def _plain(s):
    return re.sub(r"\x1b\[[0-9;]*m", "", s)


@retry_flaky
def test_elastic_grow_2_to_3(port_block):
    r = run_watch([
        "-np", "2", "-w", "-port", str(port_block), "-port-range",
        str(port_block + 2), "-builtin-config-port", str(port_block + 1),
        sys.executable, "examples/elastic_trainer.py",
        "--schedule", "3:3", "--max-step", "8",
    ])
    out = _plain(r.stdout)
    assert r.returncode == 0, out + _plain(r.stderr)
    assert out.count("JOIN") == 3  # 2 initial + 1 joiner
    resized = [ln for ln in out.splitlines() if "RESIZED" in ln]
    assert any("size=3" in ln for ln in resized)
    done = [ln for ln in out.splitlines() if "DONE" in ln]
    assert len(done) == 3
    assert all("size=3 step=8" in ln for ln in done)


@retry_flaky
def test_elastic_shrink_3_to_2(port_block):
    r = run_watch([
        "-np", "3", "-w", "-port", str(port_block), "-port-range",
        str(port_block + 2), "-builtin-config-port", str(port_block + 1),
        sys.executable, "examples/elastic_trainer.py",
        "--schedule", "2:2", "--max-step", "6",
    ])
    out = _plain(r.stdout)
    assert r.returncode == 0, out + _plain(r.stderr)
    assert out.count("DETACHED") == 1
    done = [ln for ln in out.splitlines() if "DONE" in ln]
    assert len(done) == 2
    assert all("size=2 step=6" in ln for ln in done)


@retry_flaky
def test_elastic_grow_and_shrink(port_block):
    r = run_watch([
        "-np", "2", "-w", "-port", str(port_block), "-port-range",
        str(port_block + 2), "-builtin-config-port", str(port_block + 1),
        sys.executable, "examples/elastic_trainer.py",
        "--schedule", "2:4,5:2", "--max-step", "8",
    ], timeout=300)
    out = _plain(r.stdout)
    assert r.returncode == 0, out + _plain(r.stderr)
    done = [ln for ln in out.splitlines() if "DONE" in ln]
    assert len(done) == 2
    assert all("size=2 step=8" in ln for ln in done)
    assert out.count("DETACHED") == 2


@retry_flaky
def test_elastic_churn(port_block):
    """Repeated grow/shrink churn: 2 -> 4 -> 2 -> 3 -> 2 in one run."""
    r = run_watch([
        "-np", "2", "-w", "-port", str(port_block), "-port-range",
        str(port_block + 2), "-builtin-config-port", str(port_block + 1),
        sys.executable, "examples/elastic_trainer.py",
        "--schedule", "2:4,4:2,6:3,8:2", "--max-step", "10",
    ], timeout=300)
    out = _plain(r.stdout)
    assert r.returncode == 0, out + _plain(r.stderr)
    done = [ln for ln in out.splitlines() if "DONE" in ln]
    assert len(done) == 2, out
    assert all("size=2 step=10" in ln for ln in done)
    assert out.count("RESIZED") >= 6  # at least surviving workers log each


@retry_flaky
def test_elastic_remove_rank0(port_block):
    """Removing rank 0 mid-run: the old rank 0 notifies the runners, then
    detaches; the old rank 1 becomes the new rank 0 and training continues
    (reference peer.go:214-224 'detached if self not in new cluster')."""
    script = r"""
import os
import sys

sys.path.insert(0, os.getcwd())
import torch
import kungfu_amd as kf
from examples.elastic_trainer import replace_cluster
from kungfu_amd.models import SLP
from kungfu_amd.ops import broadcast_model
from kungfu_amd.optimizers import SynchronousSGDOptimizer

kf.init(with_torch=False)
torch.manual_seed(0)
model = SLP(in_features=8, classes=2)
step = kf.all_reduce_int_max(0)
broadcast_model(model)
opt = SynchronousSGDOptimizer(torch.optim.SGD(model.parameters(), lr=0.1))
print("JOIN rank=%d size=%d" % (kf.rank(), kf.size()), flush=True)
while step < 6:
    x = torch.randn(4, 1, 2, 4)
    y = torch.randint(0, 2, (4,))
    opt.zero_grad()
    torch.nn.functional.cross_entropy(model(x), y).backward()
    opt.step()
    step += 1
    if step == 3 and kf.rank() == 0:
        peers = os.environ["KUNGFU_INIT_PEERS"].split(",")
        replace_cluster(peers[1:])  # drop myself (rank 0)
    changed, detached = kf.resize()
    if detached:
        print("DETACHED old-rank0", flush=True)
        break
    if changed:
        step = kf.all_reduce_int_max(step)
        broadcast_model(model)
        opt = SynchronousSGDOptimizer(
            torch.optim.SGD(model.parameters(), lr=0.1))
        print("RESIZED rank=%d size=%d" % (kf.rank(), kf.size()),
              flush=True)
if not kf.detached():
    print("DONE rank=%d size=%d step=%d" % (kf.rank(), kf.size(), step),
          flush=True)
kf.finalize()
"""
    r = run_watch([
        "-np", "3", "-w", "-port", str(port_block), "-port-range",
        str(port_block + 2), "-builtin-config-port", str(port_block + 1),
        sys.executable, "-c", script,
    ])
    out = _plain(r.stdout)
    assert r.returncode == 0, out + _plain(r.stderr)
    assert out.count("DETACHED old-rank0") == 1
    done = [ln for ln in out.splitlines() if "DONE" in ln]
    assert len(done) == 2, out
    assert all("size=2 step=6" in ln for ln in done)
    assert "DONE rank=0" in out and "DONE rank=1" in out  # re-ranked
