"""The SLURM launch kit (reference stage 5):
syntax-check the sbatch script and hold its torchrun invocation to the
reference contract (with the reference's line-19 continuation bug fixed)."""

import os
import subprocess

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
SBATCH = os.path.join(ROOT, "slurm", "sbatch_run.sh")


def test_sbatch_script_is_valid_bash():
    r = subprocess.run(["bash", "-n", SBATCH], capture_output=True, text=True)
    assert r.returncode == 0, r.stderr


def test_sbatch_contract():
    src = open(SBATCH).read()
    # reference contract (ref slurm/sbatch_run.sh:17-23): c10d rendezvous
    # at the head node, 4 nodes x 1 proc, multinode entrypoint, args 50 10
    assert "torchrun" in src
    assert "--rdzv_backend" in src and "c10d" in src
    assert "--rdzv_endpoint" in src and "head_node_ip" in src
    assert "multinode_torchrun.py" in src
    assert "--nnodes" in src
    # the reference's broken trailing comment after a line continuation
    # (ref :19) must NOT be reproduced
    for line in src.splitlines():
        if line.rstrip().endswith("\\"):
            continue
        assert "\\  #" not in line


def test_cluster_config_template_shape():
    import yaml
    cfg = yaml.safe_load(open(os.path.join(ROOT, "slurm",
                                           "config.yaml.template")))
    assert cfg  # parses as YAML and is non-empty
