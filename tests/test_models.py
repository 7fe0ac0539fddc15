"""Model-zoo tests (CPU): local definitions match the reference's imports
(torchvision resnet50 / vit_l_32 param census) and train end-to-end."""

import torch

from mi355x_ddp.models import resnet50, vit_l_32, vit_tiny


def test_resnet50_param_census():
    n = sum(p.numel() for p in resnet50().parameters())
    assert n == 25_557_032  # torchvision resnet50


def test_vit_l_32_param_census():
    n = sum(p.numel() for p in vit_l_32().parameters())
    assert n == 306_535_400  # torchvision vit_l_32


def test_vit_tiny_trains_cpu():
    torch.manual_seed(0)
    m = vit_tiny()
    opt = torch.optim.SGD(m.parameters(), lr=0.05)
    x = torch.rand(4, 3, 64, 64)
    t = torch.randint(0, 10, (4,))
    losses = []
    for _ in range(3):
        opt.zero_grad()
        loss = torch.nn.functional.cross_entropy(m(x), t)
        loss.backward()
        opt.step()
        losses.append(float(loss.detach()))
    assert all(torch.isfinite(torch.tensor(losses)))
    assert losses[-1] < losses[0]


def test_hip_linear_nd_input_cpu():
    # nn.Linear drop-in for arbitrary leading dims (transformer-style
    # [B, T, K] inputs) — CPU path
    import torch

    from mi355x_ddp.models.toy import HipLinear

    torch.manual_seed(4)
    m = HipLinear(20, 3)
    x = torch.randn(2, 5, 20)
    y = m(x)
    assert y.shape == (2, 5, 3)
    ref = torch.nn.functional.linear(x, m.weight, m.bias)
    assert torch.allclose(y, ref, atol=1e-6)
