"""The driver launches bench.py via torch.distributed.run — verify the
RANK/WORLD_SIZE env-synthesis path (no kungfu-run) end to end on CPU."""
import os
import re
import subprocess
import sys

from mp_helpers import retry_flaky

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


@retry_flaky
def test_bench_slp_under_torchrun_env(port_block):
    procs = []
    for rank in range(2):
        env = dict(os.environ)
        env.update({
            "PYTHONPATH": ROOT,
            "RANK": str(rank),
            "LOCAL_RANK": str(rank),
            "WORLD_SIZE": "2",
            "MASTER_ADDR": "127.0.0.1",
            "MASTER_PORT": str(port_block + 40),
            "KUNGFU_PORT_BASE": str(port_block),
        })
        procs.append(subprocess.Popen(
            [sys.executable, "bench.py", "--model", "slp", "--steps", "4",
             "--warmup", "1", "--dtype", "fp32"],
            cwd=ROOT, env=env, stdout=subprocess.PIPE,
            stderr=subprocess.PIPE, text=True))
    outs = []
    for p in procs:
        out, err = p.communicate(timeout=180)
        assert p.returncode == 0, out + err
        outs.append(out)
    # exactly one rank (rank 0) prints the JSON line with n_gpus=2
    jsons = [ln for out in outs for ln in out.splitlines()
             if ln.startswith("{")]
    assert len(jsons) == 1, outs
    assert '"n_gpus": 2' in jsons[0]


@retry_flaky
def test_elastic_trainer_class(port_block):
    env = dict(os.environ)
    env["PYTHONPATH"] = ROOT + os.pathsep + env.get("PYTHONPATH", "")
    script = r"""
import torch
import kungfu_amd as kf
from kungfu_amd.models import SLP
from kungfu_amd.optimizers import SynchronousSGDOptimizer
from kungfu_amd.parallel.elastic import ElasticTrainer

kf.init(with_torch=False)
torch.manual_seed(0)
trainer = ElasticTrainer(
    SLP(in_features=8, classes=2),
    lambda m: SynchronousSGDOptimizer(torch.optim.SGD(m.parameters(),
                                                      lr=0.1)),
    schedule="2:3")
while trainer.step < 5:
    x = torch.randn(4, 1, 2, 4)
    y = torch.randint(0, 2, (4,))
    trainer.optimizer.zero_grad()
    torch.nn.functional.cross_entropy(trainer.model(x), y).backward()
    trainer.optimizer.step()
    if not trainer.after_step():
        break
if not trainer.detached:
    print("ET-DONE size=%d step=%d" % (kf.size(), trainer.step))
kf.finalize()
"""
    r = subprocess.run(
        [sys.executable, "-m", "kungfu_amd.run", "-np", "2", "-w",
         "-port", str(port_block), "-port-range", str(port_block + 2),
         "-builtin-config-port", str(port_block + 1),
         sys.executable, "-c", script],
        cwd=ROOT, env=env, capture_output=True, text=True, timeout=240)
    out = re.sub(r"\x1b\[[0-9;]*m", "", r.stdout)
    assert r.returncode == 0, out + r.stderr
    done = [ln for ln in out.splitlines() if "ET-DONE" in ln]
    assert len(done) == 3 and all("size=3 step=5" in ln for ln in done), out


def test_process_group_env_master(port_block):
    """torchrun-style: MASTER_ADDR/PORT set -> env:// attach (gloo)."""
    from mp_helpers import dist_gloo_body, spawn_cluster

    extra = {
        "KUNGFU_TORCH_BACKEND": "gloo",
        "MASTER_ADDR": "127.0.0.1",
        "MASTER_PORT": str(port_block + 50),
    }
    res = spawn_cluster(dist_gloo_body, 2, port_block, extra=extra)
    assert res == [3.0, 3.0]


def test_process_group_derived_store(port_block):
    """kungfu-run style: no MASTER env -> rank0 hosts a derived-port
    store."""
    from mp_helpers import dist_gloo_body, spawn_cluster

    extra = {"KUNGFU_TORCH_BACKEND": "gloo"}
    res = spawn_cluster(dist_gloo_body, 2, port_block, extra=extra)
    assert res == [3.0, 3.0]


def test_pair_averaging_rccl_exchange(port_block):
    """Symmetric tournament gossip over torch.distributed sendrecv
    (gloo on CPU; RCCL on GPU boxes). With lr=0, after meeting every
    other rank the replicas must mix toward the same average."""
    from mp_helpers import pair_rccl_body, spawn_cluster

    res = spawn_cluster(pair_rccl_body, 2, port_block,
                        extra={"KUNGFU_TORCH_BACKEND": "gloo"})
    assert all(r["changed"] for r in res)
    # n=2: one round of pairwise averaging makes replicas identical
    assert res[0]["w"] == res[1]["w"]


def test_tournament_partner_schedule():
    from kungfu_amd.optimizers.async_sgd import tournament_partner

    for n in (2, 3, 4, 5, 8):
        for s in range(2 * n):
            seen = {}
            for r in range(n):
                p = tournament_partner(r, s, n)
                seen[r] = p
            for r, p in seen.items():
                if p >= 0:
                    assert seen[p] == r, (n, s, r, p, seen)  # symmetric
            idle = [r for r, p in seen.items() if p < 0]
            assert len(idle) == (n % 2)
        # over n-1 rounds (even n) every rank meets every other rank
        if n % 2 == 0:
            for r in range(n):
                met = {tournament_partner(r, s, n) for s in range(n - 1)}
                assert met == set(range(n)) - {r}, (n, r, met)


@retry_flaky
def test_bench_resnet50_n2_cpu_orchestration(port_block):
    """The FULL flagship orchestration at N=2 on CPU (driver-style env:
    torchrun synthesis, model broadcast, bucketed gradient all-reduce
    through the C++ engine, max-over-ranks timing, single JSON line) —
    everything the 8-GPU run does except the RCCL transport itself."""
    procs = []
    for rank in range(2):
        env = dict(os.environ)
        env.update({
            "PYTHONPATH": ROOT,
            "RANK": str(rank),
            "LOCAL_RANK": str(rank),
            "WORLD_SIZE": "2",
            "MASTER_ADDR": "127.0.0.1",
            "MASTER_PORT": str(port_block + 41),
            "KUNGFU_PORT_BASE": str(port_block),
        })
        procs.append(subprocess.Popen(
            [sys.executable, "bench.py", "--model", "resnet50",
             "--batch-size", "2", "--steps", "2", "--warmup", "1",
             "--dtype", "fp32", "--fused-bn", "0", "--fused-opt", "0",
             "--graph", "0"],
            cwd=ROOT, env=env, stdout=subprocess.PIPE,
            stderr=subprocess.PIPE, text=True))
    outs = []
    for p in procs:
        out, err = p.communicate(timeout=300)
        assert p.returncode == 0, out + err
        outs.append(out)
    jsons = [ln for out in outs for ln in out.splitlines()
             if ln.startswith("{")]
    assert len(jsons) == 1, outs
    assert '"n_gpus": 2' in jsons[0] and '"global_batch": 4' in jsons[0]
