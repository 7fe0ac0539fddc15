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


def test_shards_for_block_matches_per_epoch_concat():
    # CPU fallback of the multi-epoch gather: block e of shards_for_block
    # must equal _gather(e0+e) exactly (the GPU kernel path is held to the
    # same contract bitwise in tests/test_epoch_block_gpu.py)
    import torch

    sys.path.insert(0, ROOT)
    import bench

    d = bench.DeviceData(256, rank=1, world=2, batch=8,
                         device=torch.device("cpu"))
    xs_b, ts_b = d.shards_for_block(3, 4)
    per = d.per_rank
    assert xs_b.shape == (4 * per, 20) and ts_b.shape == (4 * per, 1)
    for e in range(4):
        xs, ts = d._gather(3 + e)
        assert torch.equal(xs_b[e * per:(e + 1) * per], xs)
        assert torch.equal(ts_b[e * per:(e + 1) * per], ts)
