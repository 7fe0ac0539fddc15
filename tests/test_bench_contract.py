"""The driver depends on bench.py's CLI + JSON contract; hold it stable."""

import json
import os
import subprocess
import sys

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_bench_json_contract(tmp_path):
    out = subprocess.run(
        [sys.executable, os.path.join(ROOT, "bench.py"), "--steps", "30",
         "--warmup", "5", "--p50-probes", "4"],
        cwd=ROOT, capture_output=True, text=True, timeout=300)
    assert out.returncode == 0, out.stderr[-2000:]
    r = json.loads(out.stdout.strip().splitlines()[-1])
    for key in ("metric", "value", "unit", "n_gpus", "steps", "warmup",
                "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
                "dtype", "data", "config"):
        assert key in r, key
    assert r["metric"].startswith("samples/sec")
    assert r["unit"] == "samples/s"
    assert r["steps"] == 30 and r["warmup"] == 5
    assert r["higher_is_better"] is True and r["scaling"] == "weak"
    assert r["data"] == "synthetic"
    cfg = r["config"]
    for key in ("model", "global_batch", "parallelism", "comm", "engine"):
        assert key in cfg, key
    assert cfg["model"] == "Linear(20,1)"
    assert r["value"] > 0 and r["ms_per_step"] > 0


def test_bench_defaults_finish_quickly():
    # "with no flags it must default to N=1 and a K/W that finish within
    # minutes" — run the real defaults end to end
    out = subprocess.run([sys.executable, os.path.join(ROOT, "bench.py")],
                         cwd=ROOT, capture_output=True, text=True,
                         timeout=420)
    assert out.returncode == 0, out.stderr[-2000:]
    r = json.loads(out.stdout.strip().splitlines()[-1])
    assert r["steps"] == 2000
