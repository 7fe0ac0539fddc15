"""GPU bf16 tests: bf16 storage kernels (f32-MFMA compute), the standalone
bf16 MFMA GEMM (v_mfma_f32_16x16x32_bf16), and the bf16 toy engine
(BASELINE.json config 2 capability)."""

import pytest
import torch

from mi355x_ddp import ops

pytestmark = pytest.mark.gpu

DEV = "cuda:0"


def _rand(*shape, seed=0):
    g = torch.Generator().manual_seed(seed)
    return torch.rand(*shape, generator=g)


@pytest.mark.parametrize("B,K,N", [(32, 20, 1), (33, 21, 7), (64, 64, 48)])
def test_linear_fwd_bf16(B, K, N):
    x = _rand(B, K, seed=1).to(DEV).bfloat16()
    w = (_rand(N, K, seed=2) - 0.3).to(DEV).bfloat16()
    b = _rand(N, seed=3).to(DEV).bfloat16()
    y = ops.ext().linear_fwd(x, w, b)
    assert y.dtype == torch.bfloat16
    ref = torch.nn.functional.linear(x.cpu().float(), w.cpu().float(),
                                     b.cpu().float())
    assert torch.allclose(y.cpu().float(), ref, atol=3e-2, rtol=3e-2)


@pytest.mark.parametrize("B,K,N", [(32, 32, 16), (64, 100, 48), (16, 20, 1),
                                   (128, 256, 256)])
def test_gemm_bf16_mfma(B, K, N):
    # asymmetric operands: catches transposed fragment layouts (guide G9)
    x = (_rand(B, K, seed=4) - 0.5).to(DEV).bfloat16()
    w = (_rand(N, K, seed=5) * torch.linspace(0.5, 1.5, K)).to(DEV).bfloat16()
    b = _rand(N, seed=6).to(DEV).bfloat16()
    y = ops.ext().gemm_bf16(x, w, b)
    ref = torch.nn.functional.linear(x.cpu().float(), w.cpu().float(),
                                     b.cpu().float())
    assert torch.allclose(y.cpu().float(), ref, atol=K * 2e-3, rtol=3e-2), \
        (y.cpu().float() - ref).abs().max()


def test_sgd_flat_bf16():
    p = _rand(512, seed=7).to(DEV).bfloat16()
    g = (_rand(512, seed=8) - 0.5).to(DEV).bfloat16()
    p0 = p.clone()
    g0 = g.clone()
    ops.sgd_flat_(p, g, lr=0.1, zero_grad=True)
    ref = (p0.float() - 0.1 * g0.float()).bfloat16()
    assert torch.equal(p, ref)
    assert g.float().abs().sum() == 0


def test_mse_bf16():
    y = (_rand(32, 4, seed=9) - 0.5).to(DEV).bfloat16().requires_grad_(True)
    t = _rand(32, 4, seed=10).to(DEV)
    loss = ops.mse_loss(y, t)
    loss.backward()
    yc = y.detach().cpu().float().requires_grad_(True)
    ref = torch.nn.functional.mse_loss(yc, t.cpu().float().bfloat16().float())
    ref.backward()
    assert torch.allclose(loss.cpu(), ref, atol=1e-2, rtol=1e-2)
    assert torch.allclose(y.grad.cpu().float(), yc.grad, atol=1e-2, rtol=5e-2)


def test_toy_engine_bf16_end_to_end():
    from mi355x_ddp.engine import ToyFusedStep
    from mi355x_ddp.models import toy_model
    torch.manual_seed(3)
    model = toy_model(20, 1).to(DEV).bfloat16()
    eng = ToyFusedStep(model, comm=None, lr=0.05, use_mse=True)
    X = _rand(10, 32, 20, seed=11).to(DEV).bfloat16()
    T = _rand(10, 32, 1, seed=12).to(DEV).bfloat16()
    w0 = model.weight.detach().float().cpu().clone()
    for s in range(10):
        eng.step(X[s], T[s])
    torch.cuda.synchronize()
    w1 = model.weight.detach().float().cpu()
    assert not torch.equal(w0, w1)  # actually trained
    assert torch.isfinite(w1).all()

    # numerics: compare one step against f32 reference from same start
    model2 = toy_model(20, 1).to(DEV).bfloat16()
    with torch.no_grad():
        model2.weight.copy_(torch.zeros_like(model2.weight))
        model2.bias.zero_()
    eng2 = ToyFusedStep(model2, comm=None, lr=0.1, use_mse=True)
    eng2.step(X[0], T[0])
    torch.cuda.synchronize()
    x = X[0].float().cpu()
    t = T[0].float().cpu()
    w = torch.zeros(1, 20, requires_grad=True)
    b = torch.zeros(1, requires_grad=True)
    loss = torch.nn.functional.mse_loss(
        torch.nn.functional.linear(x, w, b), t)
    loss.backward()
    assert torch.allclose(model2.weight.float().cpu(),
                          (-0.1 * w.grad), atol=2e-3, rtol=5e-2)
