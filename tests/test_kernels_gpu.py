"""GPU numerics tests: every HIP kernel vs a plain PyTorch fp32 reference
(run on the MI355X box via gpurun; SURVEY §4 consequence (b)).

Inputs are asymmetric random matrices so output/operand transposes cannot
pass (cdna_hip_programming.md §5.4 rule 16)."""

import pytest
import torch

from mi355x_ddp import ops

pytestmark = pytest.mark.gpu

DEV = "cuda:0"


def _rand(*shape, seed=0):
    g = torch.Generator().manual_seed(seed)
    return torch.rand(*shape, generator=g)


def test_ext_loads_natively():
    assert ops.has_ext(), "native extension must be present on a GPU host"
    import mi355x_ddp._C as C
    assert "_C" in C.__file__ and "mi355x_ddp" in C.__file__


@pytest.mark.parametrize("B,K,N", [(32, 20, 1), (32, 20, 16), (33, 21, 7),
                                   (128, 64, 48), (256, 100, 1000)])
def test_linear_fwd(B, K, N):
    x = _rand(B, K, seed=1).to(DEV)
    w = _rand(N, K, seed=2).to(DEV) - 0.3
    b = _rand(N, seed=3).to(DEV)
    y = ops.ext().linear_fwd(x, w, b)
    ref = torch.nn.functional.linear(x.cpu(), w.cpu(), b.cpu())
    assert torch.allclose(y.cpu(), ref, atol=1e-4, rtol=1e-4)


@pytest.mark.parametrize("B,K,N", [(32, 20, 1), (33, 21, 7), (128, 64, 48)])
def test_linear_bwd_weight(B, K, N):
    x = _rand(B, K, seed=4).to(DEV)
    dy = (_rand(B, N, seed=5) - 0.5).to(DEV)
    dw = torch.zeros(N, K, device=DEV)
    db = torch.zeros(N, device=DEV)
    ops.ext().linear_bwd_weight(x, dy, dw, db, False)
    ref_w = dy.cpu().t() @ x.cpu()
    ref_b = dy.cpu().sum(0)
    assert torch.allclose(dw.cpu(), ref_w, atol=1e-4, rtol=1e-4)
    assert torch.allclose(db.cpu(), ref_b, atol=1e-4, rtol=1e-4)
    # accumulate mode adds
    ops.ext().linear_bwd_weight(x, dy, dw, db, True)
    assert torch.allclose(dw.cpu(), 2 * ref_w, atol=1e-4, rtol=1e-4)
    assert torch.allclose(db.cpu(), 2 * ref_b, atol=1e-4, rtol=1e-4)


@pytest.mark.parametrize("B,K,N", [(32, 20, 1), (33, 21, 7), (64, 128, 32)])
def test_linear_bwd_input(B, K, N):
    dy = (_rand(B, N, seed=6) - 0.5).to(DEV)
    w = (_rand(N, K, seed=7) - 0.5).to(DEV)
    dx = ops.ext().linear_bwd_input(dy, w)
    ref = dy.cpu() @ w.cpu()
    assert torch.allclose(dx.cpu(), ref, atol=1e-4, rtol=1e-4)


def test_linear_autograd_end_to_end():
    x = _rand(32, 20, seed=8).to(DEV).requires_grad_(True)
    w = (_rand(4, 20, seed=9) - 0.5).to(DEV).requires_grad_(True)
    b = _rand(4, seed=10).to(DEV).requires_grad_(True)
    y = ops.linear(x, w, b)
    loss = (y * (_rand(32, 4, seed=11).to(DEV))).sum()
    loss.backward()

    xc = x.detach().cpu().requires_grad_(True)
    wc = w.detach().cpu().requires_grad_(True)
    bc = b.detach().cpu().requires_grad_(True)
    yc = torch.nn.functional.linear(xc, wc, bc)
    (yc * _rand(32, 4, seed=11)).sum().backward()
    assert torch.allclose(x.grad.cpu(), xc.grad, atol=1e-4, rtol=1e-4)
    assert torch.allclose(w.grad.cpu(), wc.grad, atol=1e-4, rtol=1e-4)
    assert torch.allclose(b.grad.cpu(), bc.grad, atol=1e-4, rtol=1e-4)


@pytest.mark.parametrize("B,C", [(32, 1), (32, 10), (8, 1000), (5, 3)])
def test_cross_entropy(B, C):
    y = (_rand(B, C, seed=12) * 4 - 2).to(DEV).requires_grad_(True)
    t = _rand(B, C, seed=13).to(DEV)
    loss = ops.cross_entropy(y, t)
    loss.backward()

    yc = y.detach().cpu().requires_grad_(True)
    ref = torch.nn.CrossEntropyLoss()(yc, t.cpu())
    ref.backward()
    assert torch.allclose(loss.cpu(), ref, atol=1e-5, rtol=1e-5)
    assert torch.allclose(y.grad.cpu(), yc.grad, atol=1e-5, rtol=1e-4)


@pytest.mark.parametrize("B,C", [(32, 1), (16, 33)])
def test_mse(B, C):
    y = (_rand(B, C, seed=14) - 0.5).to(DEV).requires_grad_(True)
    t = _rand(B, C, seed=15).to(DEV)
    loss = ops.mse_loss(y, t)
    loss.backward()

    yc = y.detach().cpu().requires_grad_(True)
    ref = torch.nn.functional.mse_loss(yc, t.cpu())
    ref.backward()
    assert torch.allclose(loss.cpu(), ref, atol=1e-6)
    assert torch.allclose(y.grad.cpu(), yc.grad, atol=1e-6)


def test_sgd_flat():
    p = _rand(1024, seed=16).to(DEV)
    g = (_rand(1024, seed=17) - 0.5).to(DEV)
    p0, g0 = p.clone(), g.clone()
    ops.sgd_flat_(p, g, lr=0.1, zero_grad=True)
    assert torch.allclose(p.cpu(), (p0 - 0.1 * g0).cpu(), atol=1e-7)
    assert g.abs().sum() == 0


def test_flatten_unflatten_roundtrip():
    shapes = [(3, 5), (17,), (4, 4, 4), (1,)]
    tensors = [(_rand(*s, seed=20 + i) - 0.5).to(DEV) for i, s in enumerate(shapes)]
    offsets, off = [], 0
    for t in tensors:
        offsets.append(off)
        off += (t.numel() + 3) & ~3
    total = (off + 3) & ~3
    bucket = torch.zeros(total, device=DEV)
    plan = ops.ext().build_copy_plan(tensors, offsets, 0)
    ops.ext().flatten_into(bucket, plan, plan.shape[0], False)
    for t, o in zip(tensors, offsets):
        assert torch.equal(bucket[o:o + t.numel()].view_as(t), t)
    # unflatten restores into (zeroed) sources
    originals = [t.clone() for t in tensors]
    for t in tensors:
        t.zero_()
    ops.ext().unflatten_from(bucket, plan, plan.shape[0])
    for t, o in zip(tensors, originals):
        assert torch.equal(t, o)
    # flatten with zero_src clears the sources
    ops.ext().flatten_into(bucket, plan, plan.shape[0], True)
    for t in tensors:
        assert t.abs().sum() == 0


def test_toy_fused_step_matches_reference():
    B, K = 32, 20
    x = _rand(B, K, seed=30).to(DEV)
    t = _rand(B, 1, seed=31).to(DEV)
    w = (_rand(1, K, seed=32) - 0.5).to(DEV)
    b = _rand(1, seed=33).to(DEV)
    param = torch.zeros(24, device=DEV)  # [w(20) | b | pad] padded to 4
    param[:K] = w.flatten()
    param[K] = b[0]
    grad = torch.zeros(24, device=DEV)
    loss_out = torch.zeros((), device=DEV)
    ops.ext().toy_fused_fwd_bwd(x, t, param, grad, loss_out, True, 0, K, 0.0)

    wc = w.cpu().requires_grad_(True)
    bc = b.cpu().requires_grad_(True)
    y = torch.nn.functional.linear(x.cpu(), wc, bc)
    loss = torch.nn.functional.mse_loss(y, t.cpu())
    loss.backward()
    assert torch.allclose(loss_out.cpu(), loss, atol=1e-5)
    assert torch.allclose(grad[:K].cpu(), wc.grad.flatten(), atol=1e-5)
    assert torch.allclose(grad[K].cpu(), bc.grad[0], atol=1e-5)


def test_toy_fused_inkernel_sgd():
    # lr > 0 path: one launch does fwd+bwd+SGD; matches torch step exactly
    B, K, LR = 32, 20, 0.1
    x = _rand(B, K, seed=40).to(DEV)
    t = _rand(B, 1, seed=41).to(DEV)
    w = (_rand(1, K, seed=42) - 0.5).to(DEV)
    b = _rand(1, seed=43).to(DEV)
    param = torch.zeros(24, device=DEV)
    param[:K] = w.flatten()
    param[K] = b[0]
    grad = torch.zeros(24, device=DEV)
    ops.ext().toy_fused_fwd_bwd(x, t, param, grad, torch.Tensor(), True,
                                0, K, LR)
    wc = w.cpu().requires_grad_(True)
    bc = b.cpu().requires_grad_(True)
    loss = torch.nn.functional.mse_loss(
        torch.nn.functional.linear(x.cpu(), wc, bc), t.cpu())
    loss.backward()
    assert torch.allclose(param[:K].cpu(), (wc - LR * wc.grad).detach().flatten(),
                          atol=1e-5)
    assert torch.allclose(param[K].cpu(), (bc - LR * bc.grad).detach()[0],
                          atol=1e-5)
    assert grad.abs().sum() == 0  # grads never materialized


def test_rccl_world1_comm():
    # exercises the native RCCL path end-to-end at world 1 (uid creation,
    # comm init, comm-stream all-reduce, event fencing, broadcast, barrier)
    C = ops.ext()
    uid = C.RcclComm.make_unique_id()
    comm = C.RcclComm(uid, 0, 1, 0)
    t = _rand(1024, seed=50).to(DEV)
    ref = t.clone()
    comm.all_reduce_avg(t)   # avg over world 1 == identity
    comm.join_compute()
    torch.cuda.synchronize()
    assert torch.allclose(t, ref)
    comm.all_reduce_avg_inline(t)
    comm.broadcast(t, 0)
    comm.barrier()
    assert torch.allclose(t, ref)
    del comm


def test_single_gpu_trainer_runs(tmp_chdir):
    from mi355x_ddp.data import ToyDataset, prepare_dataloader
    from mi355x_ddp.models import toy_model
    from mi355x_ddp.parallel import FusedSGD
    from mi355x_ddp.trainer import Trainer
    ds = ToyDataset(64, seed=0)
    model = toy_model(20, 1)
    opt = FusedSGD(model.parameters(), lr=1e-3)
    tr = Trainer(model, prepare_dataloader(ds, 32), opt, 0, 1,
                 loss_fn="mse", wrap_ddp=False)
    tr.train(2)
    import os
    assert os.path.exists("checkpoint.pt")


def test_epoch_shard_matches_python_mirror():
    # kernel permutation == ops.perm_index mirror, and the world's shards
    # together form an exact permutation of the dataset rows
    n, K, world = 2048, 20, 4
    g = torch.Generator().manual_seed(5)
    X = torch.rand(n, K, generator=g).to(DEV)
    T = torch.rand(n, 1, generator=g).to(DEV)
    seed = 1007
    seen = []
    per = n // world
    for rank in range(world):
        xs, ts = ops.ext().epoch_shard(X, T, seed, rank, world)
        idx = [ops.perm_index(seed, rank + i * world, n) for i in range(per)]
        seen += idx
        ref_x = X[torch.tensor(idx, device=DEV)]
        ref_t = T[torch.tensor(idx, device=DEV)]
        assert torch.equal(xs, ref_x)
        assert torch.equal(ts, ref_t)
    assert sorted(seen) == list(range(n))  # exact permutation, no repeats


def test_epoch_shard_different_epochs_differ():
    n, K = 256, 20
    X = torch.rand(n, K).to(DEV)
    T = torch.rand(n, 1).to(DEV)
    a, _ = ops.ext().epoch_shard(X, T, 1, 0, 1)
    b, _ = ops.ext().epoch_shard(X, T, 2, 0, 1)
    assert not torch.equal(a, b)


def test_ce_class_index_targets_gpu():
    """Index-target CE on device: loss + input grads match torch's
    index-target CE (the kernels see the exact one-hot equivalent)."""
    from mi355x_ddp import ops

    torch.manual_seed(9)
    y = torch.randn(16, 50, device="cuda", requires_grad=True)
    idx = torch.randint(0, 50, (16,), device="cuda")
    loss = ops.cross_entropy(y, idx)
    want = torch.nn.functional.cross_entropy(y.detach(), idx)
    assert torch.allclose(loss, want, atol=1e-5)
    loss.backward()
    y2 = y.detach().clone().requires_grad_(True)
    torch.nn.functional.cross_entropy(y2, idx).backward()
    assert torch.allclose(y.grad, y2.grad, atol=1e-5)


def test_linear_nd_leading_dims():
    """HipLinear accepts [B, T, K] like nn.Linear: forward and grads match
    torch (flatten-to-2D around the MFMA kernels)."""
    from mi355x_ddp import ops

    torch.manual_seed(12)
    x = torch.randn(4, 7, 20, device="cuda", requires_grad=True)
    w = torch.randn(3, 20, device="cuda", requires_grad=True)
    b = torch.randn(3, device="cuda", requires_grad=True)
    y = ops.linear(x, w, b)
    assert y.shape == (4, 7, 3)
    y.sum().backward()
    x2 = x.detach().clone().requires_grad_(True)
    w2 = w.detach().clone().requires_grad_(True)
    b2 = b.detach().clone().requires_grad_(True)
    torch.nn.functional.linear(x2, w2, b2).sum().backward()
    for a, r in ((x.grad, x2.grad), (w.grad, w2.grad), (b.grad, b2.grad)):
        assert torch.allclose(a, r, atol=1e-4, rtol=1e-4)
