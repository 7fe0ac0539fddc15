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


def _free_port():
    import socket
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return str(s.getsockname()[1])


def test_multigpu_spawn_world2_one_device(tmp_path):
    """Pre-flight for the driver's 8-GPU run: the
    mp.spawn entrypoint at world 2 with both ranks on ONE device. RCCL
    refuses same-device ranks, so this also proves the entrypoint path's
    all-ranks-agreed downgrade to gloo (create_comm -> build_gpu_comm)."""
    r = _run([sys.executable, os.path.join(ROOT, "multigpu.py"), "2", "1"],
             cwd=tmp_path,
             env_extra={"MI355X_FORCE_DEV0": "1", "MI355X_WORLD": "2",
                        "MASTER_PORT": _free_port()},
             timeout=600)
    assert r.returncode == 0, r.stderr[-3000:]
    assert "Epoch 1" in r.stdout
    assert (tmp_path / "checkpoint.pt").exists()


def test_torchrun_world2_one_device_snapshot(tmp_path):
    """Stage 3 (torchrun) at world 2 on one device — the exact launch form
    the driver uses at N>1, rehearsed on a 1-GPU box."""
    script = os.path.join(ROOT, "multigpu_torchrun.py")
    base = TORCHRUN + ["--standalone", "--local-addr", "127.0.0.1",
                       "--nproc_per_node", "2", script]
    r = _run(base + ["2", "1"], cwd=tmp_path,
             env_extra={"MI355X_FORCE_DEV0": "1"}, timeout=600)
    assert r.returncode == 0, r.stderr[-3000:]
    assert (tmp_path / "snapshot.pt").exists()


def test_profile_stage_world2_one_device(tmp_path):
    """Profile stage at world 2 on one device: per-rank TB traces AND the
    per-rank kernel summaries come out of one command."""
    r = _run([sys.executable, os.path.join(ROOT, "multigpu_profile.py"), "2"],
             cwd=tmp_path,
             env_extra={"MI355X_FORCE_DEV0": "1", "MI355X_WORLD": "2",
                        "MASTER_PORT": _free_port(),
                        "MI355X_PROFILE_DATASET": "128",
                        "MI355X_PROFILE_BATCH": "16"},
             timeout=900)
    assert r.returncode == 0, r.stderr[-3000:]
    trace_dir = tmp_path / "log" / "resnet50"
    stats = sorted(trace_dir.glob("kernel_stats_rank*.json"))
    assert len(stats) == 2, list(trace_dir.iterdir())
    import json
    rows = json.loads(stats[0].read_text())
    assert rows and any("Conv" in r["name"] or "conv" in r["name"]
                        or "igemm" in r["name"] or "miopen" in r["name"].lower()
                        for r in rows[:15]), [r["name"] for r in rows[:15]]
    traces = list(trace_dir.glob("*.pt.trace.json*")) \
        + list(trace_dir.glob("*.json.gz"))
    assert len([p for p in traces if "kernel_stats" not in p.name]) >= 2


def test_multinode_two_groups_of_four_one_device(tmp_path):
    """BASELINE.json config 5 on GPU: 2 x 4-rank torchrun groups (c10d
    rendezvous) forming world 8, all ranks time-sharing one device
    (MI355X_FORCE_DEV0). Covers the compound cpu:gloo,cuda:nccl backend,
    the agreed RCCL->gloo downgrade at world 8, and the multinode
    entrypoint's MSE/snapshot path in the driver-like launch form."""
    import subprocess
    script = os.path.join(ROOT, "multinode_torchrun.py")
    port = _free_port()
    env = dict(os.environ)
    env["PYTHONPATH"] = ROOT + os.pathsep + env.get("PYTHONPATH", "")
    env.setdefault("OMP_NUM_THREADS", "1")
    env["MI355X_FORCE_DEV0"] = "1"
    procs = []
    for node in range(2):
        procs.append(subprocess.Popen(
            TORCHRUN + ["--nnodes", "2", "--nproc_per_node", "4",
                        "--node-rank", str(node), "--local-addr", "127.0.0.1",
                        "--rdzv_backend", "c10d",
                        "--rdzv_endpoint", f"127.0.0.1:{port}",
                        "--rdzv_id", "gpujob8", script, "1", "1"],
            cwd=tmp_path, env=env, stdout=subprocess.PIPE,
            stderr=subprocess.STDOUT, text=True))
    outs = []
    for p in procs:
        out, _ = p.communicate(timeout=540)
        outs.append(out)
    assert all(p.returncode == 0 for p in procs), outs[-1][-3000:]
    allout = "\n".join(outs)
    for r in range(8):
        assert f"[GPU{r}] Epoch 0" in allout, allout[-2000:]
    assert "Steps: 8" in allout  # 2048/8 ranks/32 batch
    assert (tmp_path / "snapshot.pt").exists()
