"""The driver's contract with bench.py: one JSON line on stdout from rank
0 with the documented schema. Protects the round-end BENCH/SCALE runs
from accidental schema drift."""
import json
import os
import subprocess
import sys

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_bench_json_schema():
    env = dict(os.environ)
    env["PYTHONPATH"] = ROOT
    for k in ("RANK", "WORLD_SIZE", "LOCAL_RANK", "MASTER_ADDR",
              "MASTER_PORT", "KUNGFU_SELF_SPEC", "KUNGFU_INIT_PEERS"):
        env.pop(k, None)
    out = subprocess.run(
        [sys.executable, "bench.py", "--model", "slp", "--steps", "3",
         "--warmup", "1", "--dtype", "fp32"],
        cwd=ROOT, env=env, capture_output=True, text=True, timeout=300)
    assert out.returncode == 0, out.stdout + out.stderr
    lines = [ln for ln in out.stdout.splitlines() if ln.startswith("{")]
    assert len(lines) == 1, out.stdout
    d = json.loads(lines[0])
    for key in ("metric", "value", "unit", "n_gpus", "steps", "warmup",
                "ms_per_step", "higher_is_better", "scaling",
                "vs_baseline", "dtype", "data", "config"):
        assert key in d, key
    assert d["n_gpus"] == 1 and d["steps"] == 3 and d["warmup"] == 1
    assert d["higher_is_better"] is True
    assert d["scaling"] == "weak"
    assert d["data"] == "synthetic"
    assert d["value"] > 0 and d["ms_per_step"] > 0
    for key in ("model", "global_batch", "parallelism"):
        assert key in d["config"], key


def test_bench_default_flags_fast():
    """`python bench.py` with NO flags must default to N=1 and finish in
    minutes (driver requirement). Verified via arg defaults (the full
    default model needs a GPU)."""
    sys.path.insert(0, ROOT)
    import importlib

    bench = importlib.import_module("bench")
    old = sys.argv
    try:
        sys.argv = ["bench.py"]
        args = bench.parse_args()
    finally:
        sys.argv = old
    assert args.gpus == 1
    assert args.steps * args.batch_size <= 64 * 64  # bounded timed work
    assert args.warmup <= 15
