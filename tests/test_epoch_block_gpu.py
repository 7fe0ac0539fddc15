"""Epoch-block deferral: multi-epoch shard gather + cross-epoch multistep.

The persistent engine's deferral historically broke at every epoch
boundary (per-epoch gather launch + per-epoch multistep launch).
`epoch_shard_multi` gathers E consecutive epoch shards in ONE kernel,
bitwise-identical per block to `epoch_shard`, so bench.py can bind E
epochs at once and the multistep kernel runs E*steps_per_epoch SGD steps
per launch. These tests hold both halves to bitwise equality.
"""
import pytest
import torch

pytestmark = pytest.mark.gpu


def _ext():
    from mi355x_ddp import ops
    return ops.ext()


@pytest.mark.parametrize("world,epochs,seed0",
                         [(1, 5, 1000), (2, 3, 1500), (8, 2, 1000)])
def test_epoch_shard_multi_blocks_bitwise(world, epochs, seed0):
    torch.manual_seed(7)
    X = torch.rand(2048, 20, device="cuda")
    T = torch.rand(2048, 1, device="cuda")
    per = 2048 // world
    for rank in range(world):
        xs_m, ts_m = _ext().epoch_shard_multi(X, T, seed0, epochs, rank, world)
        assert xs_m.shape == (epochs * per, 20) and ts_m.shape == (epochs * per, 1)
        for e in range(epochs):
            xs, ts = _ext().epoch_shard(X, T, seed0 + e, rank, world)
            assert torch.equal(xs_m[e * per:(e + 1) * per], xs)
            assert torch.equal(ts_m[e * per:(e + 1) * per], ts)


@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
def test_epoch_shard_multi_dtype(dtype):
    X = torch.rand(512, 20, device="cuda").to(dtype)
    T = torch.rand(512, 1, device="cuda").to(dtype)
    xs_m, ts_m = _ext().epoch_shard_multi(X, T, 42, 3, 0, 1)
    xs1, ts1 = _ext().epoch_shard(X, T, 43, 0, 1)
    assert torch.equal(xs_m[512:1024], xs1)
    assert torch.equal(ts_m[512:1024], ts1)


def _run_epochs(X, T, block, E, spe, batch, dtype):
    from mi355x_ddp.engine import PersistentToyStep
    from mi355x_ddp.models import toy_model
    torch.manual_seed(4242)
    m = toy_model(20, 1).to(device="cuda", dtype=dtype)
    eng = PersistentToyStep(m, lr=1e-3, use_mse=True)
    if block:  # ONE bind covering E epochs -> one multistep launch
        xs, ts = _ext().epoch_shard_multi(X, T, 1000, E, 0, 1)
        eng.bind_shard(xs, ts, batch)
        for i in range(E * spe):
            eng.step_shard(i)
    else:  # the historical per-epoch path: E binds, E launches
        for e in range(E):
            xs, ts = _ext().epoch_shard(X, T, 1000 + e, 0, 1)
            eng.bind_shard(xs, ts, batch)
            for i in range(spe):
                eng.step_shard(i)
    eng.flush()
    torch.cuda.synchronize()
    return eng.flat_param.clone()


@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
def test_block_bound_run_matches_per_epoch(dtype):
    torch.manual_seed(3)
    X = torch.rand(2048, 20, device="cuda").to(dtype)
    T = torch.rand(2048, 1, device="cuda").to(dtype)
    a = _run_epochs(X, T, True, 3, 64, 32, dtype)
    b = _run_epochs(X, T, False, 3, 64, 32, dtype)
    assert torch.equal(a, b)


def test_bench_epoch_block_json():
    """bench.py default path picks an epoch block > 1 and reports it."""
    import json
    import subprocess
    import sys
    out = subprocess.run(
        [sys.executable, "bench.py", "--steps", "600", "--warmup", "100"],
        capture_output=True, text=True, timeout=300, check=True)
    line = json.loads(out.stdout.strip().splitlines()[-1])
    assert line["config"]["epoch_block"] > 1
    assert line["config"]["engine"] == "persistent"
