"""GPU entrypoint integration: the reference CLIs drive the engine on a
real MI355X (the CPU tier runs the same flows on gloo; this tier checks
the cuda/RCCL wiring end-to-end for stages 1 and 3)."""

import os
import sys

import pytest

from test_entrypoints import ROOT, TORCHRUN, _run

pytestmark = pytest.mark.gpu


def test_single_gpu_script_on_gpu(tmp_path):
    r = _run([sys.executable, os.path.join(ROOT, "single_gpu.py"), "2", "1"],
             cwd=tmp_path)
    assert r.returncode == 0, r.stderr[-2000:]
    assert "Epoch 1" in r.stdout
    assert (tmp_path / "checkpoint.pt").exists()


def test_torchrun_world1_snapshot_resume_on_gpu(tmp_path):
    script = os.path.join(ROOT, "multigpu_torchrun.py")
    base = TORCHRUN + ["--standalone", "--local-addr", "127.0.0.1",
                       "--nproc_per_node", "1", script]
    r1 = _run(base + ["2", "1"], cwd=tmp_path)
    assert r1.returncode == 0, r1.stderr[-2000:]
    assert (tmp_path / "snapshot.pt").exists()
    r2 = _run(base + ["3", "1"], cwd=tmp_path)
    assert r2.returncode == 0, r2.stderr[-2000:]
    assert "Resuming training from snapshot at Epoch" in r2.stdout
