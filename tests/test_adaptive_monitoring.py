"""Adaptive strategy switching, metrics endpoint, elastic sampler,
policy runner (reference: adaptiveStrategies.go, monitor/server.go,
datasets/adaptor.py, policy/)."""
from mp_helpers import (adaptive_body, metrics_body, retry_flaky,
                        sampler_body, spawn_cluster)


def test_interference_vote_switches_strategy(port_block):
    results = spawn_cluster(adaptive_body, 2, port_block, "STAR")
    for r in results:
        assert r["before"] == "STAR"
        assert r["after"] != "STAR"
        assert r["sum"] == 2.0
    assert results[0]["after"] == results[1]["after"]


def test_metrics_endpoint(port_block):
    results = spawn_cluster(metrics_body, 2, port_block,
                            extra={"KUNGFU_CONFIG_ENABLE_MONITORING": "1",
                                   "KUNGFU_NO_UNIX_SOCK": "1"})
    assert all(results)


def test_elastic_shard_sampler(port_block):
    a, b = spawn_cluster(sampler_body, 2, port_block)
    assert len(set(a) & set(b)) == 0
    assert len(a) + len(b) == 100


def test_policy_runner():
    from kungfu_amd.policy import BasePolicy, PolicyRunner

    calls = []

    class P(BasePolicy):
        def before_step(self):
            calls.append("bs")

        def after_step(self):
            calls.append("as")

        def after_epoch(self):
            calls.append("ae")

    r = PolicyRunner([P()], batch_size=32)
    r.before_train()
    for _ in range(2):
        r.before_step()
        r.after_step()
    r.after_epoch()
    assert calls == ["bs", "as", "bs", "as", "ae"]
    assert r.trained_samples == 64  # single process: size() == 1
    assert r.epoch == 1 and r.step == 2


def test_fakemodel_sizes():
    from kungfu_amd.models.fakemodel import model_sizes, total_params

    rs = model_sizes("resnet50-imagenet")
    assert len(rs) == 161  # canonical ResNet-50 trainable tensor count
    assert 25_500_000 < total_params("resnet50-imagenet") < 25_700_000
    assert total_params("slp-mnist") == 28 * 28 * 10 + 10
    assert 108e6 < total_params("bert") < 135e6
    assert 135e6 < total_params("vgg16-imagenet") < 140e6


@retry_flaky
def test_adaptive_example(port_block):
    import os
    import subprocess
    import sys

    ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    from mp_helpers import run_launcher_graceful

    env = dict(os.environ)
    env["PYTHONPATH"] = ROOT + os.pathsep + env.get("PYTHONPATH", "")
    rc, out, err = run_launcher_graceful(
        [sys.executable, "-m", "kungfu_amd.run", "-np", "3", "-port",
         str(port_block), "-port-range", str(port_block + 1),
         "-strategy", "STAR",
         sys.executable, "examples/adaptive_trainer.py", "--steps", "8",
         "--mst"], ROOT, env, 180)
    assert rc == 0, out + err
    assert out.count("ADAPT-DONE") == 3
    assert "MST parent array" in out
    assert "strategy 0:" in out  # stats printed


def test_egress_rates_helper(port_block):
    from mp_helpers import egress_rates_body

    assert all(spawn_cluster(egress_rates_body, 2, port_block))
