"""GPU tests for the profile-stage workload: local ResNet-50 through the
DDP engine (multi-bucket reducer + fused SGD; conv/BN via MIOpen, FC via
the MFMA linear, CE loss kernel with C=1000)."""

import pytest
import torch

from mi355x_ddp import ops
from mi355x_ddp.models import resnet50
from mi355x_ddp.parallel import FusedSGD
from mi355x_ddp.parallel.reducer import Reducer

pytestmark = pytest.mark.gpu

DEV = "cuda:0"


def test_resnet50_train_steps_multibucket():
    torch.manual_seed(0)
    model = resnet50().to(DEV)
    params = list(model.parameters())
    reducer = Reducer(params, comm=None, bucket_cap_mb=25.0)
    # ~102 MB fp32 grads -> ~5 buckets at 25 MB
    assert 4 <= len(reducer.buckets) <= 8, len(reducer.buckets)
    opt = FusedSGD(params, lr=1e-3)
    opt.attach_reducer(reducer)

    x = torch.rand(8, 3, 224, 224, device=DEV)
    t = torch.rand(8, 1000, device=DEV)
    losses = []
    for _ in range(3):
        y = model(x)
        loss = ops.cross_entropy(y, t)
        loss.backward()
        reducer.finalize()
        opt.step()
        losses.append(float(loss))
    assert all(torch.isfinite(torch.tensor(losses)))
    # same batch, 3 steps: loss should move (params actually update)
    assert losses[0] != losses[-1]


def test_resnet50_grads_match_plain_torch():
    # one backward through the engine == plain autograd grads. Both runs use
    # the framework CE kernel so the comparison isolates the reducer's
    # rebinding (flat param/grad views); tolerance is norm-relative because
    # MIOpen conv-backward uses atomics and is not bitwise repeatable.
    torch.manual_seed(1)
    model = resnet50().to(DEV)
    x = torch.rand(4, 3, 224, 224, device=DEV)
    t = torch.softmax(torch.rand(4, 1000, device=DEV), dim=1)
    bn_state = {k: v.clone() for k, v in model.state_dict().items()}

    # plain reference first (before reducer rebinding)
    y = model(x)
    loss = ops.cross_entropy(y, t)
    loss.backward()
    ref_grads = [p.grad.detach().clone() for p in model.parameters()]
    for p in model.parameters():
        p.grad = None
    model.load_state_dict(bn_state)  # rewind BN running stats

    reducer = Reducer(list(model.parameters()), comm=None, bucket_cap_mb=25.0)
    y2 = model(x)
    loss2 = ops.cross_entropy(y2, t)
    loss2.backward()
    reducer.finalize()
    assert torch.allclose(loss2, loss, atol=1e-4, rtol=1e-4)
    for p, ref in zip(model.parameters(), ref_grads):
        assert p.grad is not None
        num = (p.grad - ref).norm()
        den = ref.norm() + 1e-8
        assert num / den < 1e-2, (p.shape, float(num / den),
                                  (p.grad - ref).abs().max())
