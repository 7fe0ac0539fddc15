"""Sweep MI355X_EPOCH_BLOCK (bench epoch-block size) on one GPU."""
import json
import os
import subprocess
import sys

for eb in (2, 4, 8, 16):
    for dt in ("fp32", "bf16"):
        env = dict(os.environ, MI355X_EPOCH_BLOCK=str(eb))
        out = subprocess.run(
            [sys.executable, "bench.py", "--steps", "12000", "--warmup",
             "1000", "--dtype", dt, "--p50-probes", "0"],
            capture_output=True, text=True, timeout=90, env=env)
        d = json.loads(out.stdout.strip().splitlines()[-1])
        print(f"eb={eb:3d} {dt}: {d['value'] / 1e6:6.2f}M "
              f"{d['ms_per_step'] * 1e6:5.0f}ns", flush=True)
