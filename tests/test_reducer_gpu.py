"""GPU stress tests for the Reducer under varied bucket shapes: a toy MLP
with many mixed-size params, both gradient-transport modes, vs plain
autograd (the reducer is the one genuinely racy component)."""

import pytest
import torch
from torch import nn

from mi355x_ddp import ops
from mi355x_ddp.parallel import FusedSGD
from mi355x_ddp.parallel.reducer import Reducer

pytestmark = pytest.mark.gpu

DEV = "cuda:0"


def _mlp(seed):
    torch.manual_seed(seed)
    return nn.Sequential(
        nn.Linear(64, 512), nn.ReLU(),
        nn.Linear(512, 512), nn.ReLU(),
        nn.Linear(512, 128), nn.ReLU(),
        nn.Linear(128, 10)).to(DEV)


@pytest.mark.parametrize("grad_views", [True, False])
@pytest.mark.parametrize("cap_mb", [0.05, 0.5, 64.0])
def test_reducer_mlp_matches_autograd(grad_views, cap_mb):
    x = torch.randn(16, 64, device=DEV)
    t = torch.randn(16, 10, device=DEV)

    ref = _mlp(0)
    opt_ref = torch.optim.SGD(ref.parameters(), lr=0.01)
    for _ in range(5):
        opt_ref.zero_grad()
        torch.nn.functional.mse_loss(ref(x), t).backward()
        opt_ref.step()

    model = _mlp(0)
    red = Reducer(list(model.parameters()), comm=None,
                  bucket_cap_mb=cap_mb, grad_views=grad_views)
    # ~1.45 MB of fp32 params: small caps shard into several buckets,
    # large cap leaves first-bucket(1MB) + remainder
    assert len(red.buckets) >= (5 if cap_mb < 0.1 else
                                3 if cap_mb < 1.0 else 1)
    if cap_mb >= 1.0:
        assert len(red.buckets) <= 2
    opt = FusedSGD(model.parameters(), lr=0.01)
    opt.attach_reducer(red)
    for _ in range(5):
        torch.nn.functional.mse_loss(model(x), t).backward()
        red.finalize()
        opt.step()
    torch.cuda.synchronize()
    for p, pr in zip(model.parameters(), ref.parameters()):
        assert torch.allclose(p, pr, atol=1e-5, rtol=1e-4), \
            (p.shape, (p - pr).abs().max())


def _free_port():
    import socket
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def _ddp_gpu_worker(rank, world, port, grad_views, out):
    import os
    import torch.multiprocessing  # noqa: F401
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    torch.distributed.init_process_group("gloo", rank=rank, world_size=world)
    torch.cuda.set_device(0)
    try:
        from mi355x_ddp.parallel import DDP, FusedSGD
        from mi355x_ddp.parallel.comm import GlooComm
        model = _mlp(0)
        eng = DDP(model, comm=GlooComm(), bucket_cap_mb=0.5,
                  grad_views=grad_views)
        opt = FusedSGD(model.parameters(), lr=0.01)
        opt.attach_reducer(eng.reducer)
        g = torch.Generator().manual_seed(50 + rank)  # different data/rank
        x = torch.randn(8, 64, generator=g).to(DEV)
        t = torch.randn(8, 10, generator=g).to(DEV)
        for _ in range(4):
            torch.nn.functional.mse_loss(eng(x), t).backward()
            eng.finalize_backward()
            opt.step()
        torch.cuda.synchronize()
        if rank == 0:
            torch.save([p.detach().cpu() for p in model.parameters()], out)
        torch.distributed.barrier()
    finally:
        torch.distributed.destroy_process_group()


@pytest.mark.parametrize("grad_views", [True, False])
def test_ddp_multibucket_world2_on_device(grad_views, tmp_path):
    # distributed multi-bucket path with DEVICE tensors: 2 processes share
    # one GPU, gradient buckets all-reduced over the comm layer == a
    # single process training on the concatenated batch
    import torch.multiprocessing as mp
    out = str(tmp_path / "ddp.pt")
    mp.spawn(_ddp_gpu_worker, args=(2, _free_port(), grad_views, out),
             nprocs=2, join=True)
    got = torch.load(out, weights_only=True)

    # reference: single process, full batch (grad averaging == mean loss
    # over the union for equal-sized shards)
    ref = _mlp(0)
    opt = torch.optim.SGD(ref.parameters(), lr=0.01)
    gs = [torch.Generator().manual_seed(50 + r) for r in range(2)]
    x = torch.cat([torch.randn(8, 64, generator=g) for g in gs]).to(DEV)
    t = torch.cat([torch.randn(8, 10, generator=g) for g in gs]).to(DEV)
    for _ in range(4):
        opt.zero_grad()
        torch.nn.functional.mse_loss(ref(x), t).backward()
        opt.step()
    for p, pr in zip(got, ref.parameters()):
        assert torch.allclose(p, pr.detach().cpu(), atol=1e-5, rtol=1e-4), \
            (p - pr.detach().cpu()).abs().max()


def test_copy_mode_plan_rebuilds_after_grad_replacement():
    """Copy mode captures raw grad pointers in its flatten plan; if the
    user replaces a grad tensor between steps (p.grad = None forces
    autograd to allocate a fresh one), the plan must be rebuilt — a stale
    plan would silently flatten last iteration's memory."""
    x = torch.randn(16, 64, device=DEV)
    t = torch.randn(16, 10, device=DEV)

    ref = _mlp(3)
    opt_ref = torch.optim.SGD(ref.parameters(), lr=0.01)
    for step in range(4):
        opt_ref.zero_grad(set_to_none=True)
        torch.nn.functional.mse_loss(ref(x), t).backward()
        opt_ref.step()

    model = _mlp(3)
    red = Reducer(list(model.parameters()), comm=None,
                  bucket_cap_mb=0.5, grad_views=False)
    opt = FusedSGD(model.parameters(), lr=0.01)
    opt.attach_reducer(red)
    plans_before = None
    for step in range(4):
        if step == 2:
            # force autograd to allocate fresh grad tensors mid-run
            for p in model.parameters():
                p.grad = None
            plans_before = [b.plan for b in red.buckets]
        torch.nn.functional.mse_loss(model(x), t).backward()
        red.finalize()
        opt.step()
    torch.cuda.synchronize()
    # every plan was rebuilt after the replacement...
    assert all(b.plan is not old
               for b, old in zip(red.buckets, plans_before))
    # ...and the training history matches plain autograd
    for p, pr in zip(model.parameters(), ref.parameters()):
        assert torch.allclose(p, pr, atol=1e-5, rtol=1e-4), \
            (p.shape, (p - pr).abs().max())
