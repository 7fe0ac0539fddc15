"""GPU tests for the copy-stream H2D prefetcher and its use
in the Trainer hot loop."""

import pytest
import torch

from mi355x_ddp.data import DevicePrefetcher, ToyDataset, prepare_dataloader

pytestmark = pytest.mark.gpu

DEV = torch.device("cuda", 0)


def test_prefetcher_contents_and_device():
    ds = ToyDataset(128, seed=3)
    loader = prepare_dataloader(ds, 16, shuffle=False)
    ref = [(x.clone(), t.clone()) for x, t in loader]
    got = list(DevicePrefetcher(loader, DEV))
    torch.cuda.synchronize()
    assert len(got) == len(ref)
    for (xr, tr), (xg, tg) in zip(ref, got):
        assert xg.is_cuda and tg.is_cuda
        assert torch.equal(xg.cpu(), xr) and torch.equal(tg.cpu(), tr)


def test_trainer_epoch_with_prefetcher(tmp_path):
    from mi355x_ddp.models import toy_model
    from mi355x_ddp.trainer import Trainer
    torch.manual_seed(0)
    model = toy_model(20, 1)
    ds = ToyDataset(256, seed=1)
    loader = prepare_dataloader(ds, 32, shuffle=False)
    opt = torch.optim.SGD(model.parameters(), lr=1e-3)
    tr = Trainer(model, loader, opt, 0, save_every=10**9, loss_fn="mse",
                 wrap_ddp=False, checkpoint_path=str(tmp_path / "c.pt"))
    w0 = model.weight.detach().cpu().clone()
    tr.train(1)
    torch.cuda.synchronize()
    assert not torch.equal(model.weight.detach().cpu(), w0)
    assert torch.isfinite(model.weight.detach()).all()
