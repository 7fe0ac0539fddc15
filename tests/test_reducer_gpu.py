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
