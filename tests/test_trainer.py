"""CPU tests of the Trainer loop, checkpoint and snapshot formats
(reference behaviors from SURVEY §3.1/§3.3, §5.4)."""

import os

import torch

from mi355x_ddp.data import ToyDataset, prepare_dataloader
from mi355x_ddp.models import toy_model
from mi355x_ddp.parallel import FusedSGD
from mi355x_ddp.trainer import Trainer


def _make(snapshot_path=None, loss_fn="ce", size=64):
    ds = ToyDataset(size, seed=0)
    model = toy_model(20, 1)
    opt = FusedSGD(model.parameters(), lr=1e-3)
    dl = prepare_dataloader(ds, 32)
    return Trainer(model, dl, opt, "cpu", 1, snapshot_path=snapshot_path,
                   loss_fn=loss_fn, wrap_ddp=False)


def test_single_epoch_and_checkpoint(tmp_chdir, capsys):
    t = _make()
    t.train(1)
    out = capsys.readouterr().out
    # reference banner format (multigpu.py:46-47)
    assert "[GPUcpu] Epoch 0 | Batchsize: 32 | Steps: 2" in out
    assert os.path.exists("checkpoint.pt")
    sd = torch.load("checkpoint.pt", weights_only=True)
    assert set(sd.keys()) == {"weight", "bias"}  # raw state_dict format
    assert sd["weight"].shape == (1, 20)


def test_mse_training_reduces_loss(tmp_chdir):
    torch.manual_seed(0)
    ds = ToyDataset(256, seed=1)
    model = toy_model(20, 1)
    opt = FusedSGD(model.parameters(), lr=0.1)
    dl = prepare_dataloader(ds, 32, shuffle=False)
    tr = Trainer(model, dl, opt, "cpu", 10 ** 6, loss_fn="mse", wrap_ddp=False)

    def eval_loss():
        with torch.no_grad():
            return torch.nn.functional.mse_loss(
                model(ds.inputs), ds.targets).item()

    before = eval_loss()
    tr.train(5)
    after = eval_loss()
    assert after < before * 0.9  # actually learns (reference's CE cannot)


def test_snapshot_save_resume_format(tmp_chdir, capsys):
    snap = "snapshot.pt"
    t = _make(snapshot_path=snap)
    t.train(3)
    assert os.path.exists(snap)
    payload = torch.load(snap, weights_only=True)
    # byte-compatible reference snapshot schema (multigpu_torchrun.py:57-62)
    assert set(payload.keys()) == {"MODEL_STATE", "EPOCHS_RUN"}
    assert payload["EPOCHS_RUN"] == 2

    # a fresh trainer resumes from the snapshot epoch
    t2 = _make(snapshot_path=snap)
    out = capsys.readouterr().out
    assert "Resuming training from snapshot at Epoch 2" in out
    assert t2.epochs_run == 2
    t2.train(4)
    out = capsys.readouterr().out
    assert "Epoch 2" in out and "Epoch 3" in out and "Epoch 1" not in out


def test_ce_loss_matches_torch_semantics():
    # the toy CE is degenerate (C=1 -> loss 0): verify we match
    # torch exactly rather than inventing different semantics
    y = torch.randn(8, 1, requires_grad=True)
    t = torch.rand(8, 1)
    from mi355x_ddp import ops
    loss = ops.cross_entropy(y, t)
    ref = torch.nn.CrossEntropyLoss()(y, t)
    assert torch.allclose(loss, ref)
    loss.backward()
    assert torch.allclose(y.grad, torch.zeros_like(y.grad))


def test_ce_loss_multiclass_matches_torch():
    y = torch.randn(8, 10, requires_grad=True)
    t = torch.softmax(torch.randn(8, 10), dim=1)
    from mi355x_ddp import ops
    loss = ops.cross_entropy(y, t)
    ref = torch.nn.CrossEntropyLoss()(y.detach(), t)
    assert torch.allclose(loss, ref, atol=1e-6)


def test_trainer_fast_engine_falls_back_on_cpu(tmp_path):
    # engine="auto" on a CPU host: conditions unmet (no CUDA) -> hooks path
    import torch
    from mi355x_ddp.data import ToyDataset, prepare_dataloader
    from mi355x_ddp.models import toy_model
    from mi355x_ddp.parallel import FusedSGD
    from mi355x_ddp.trainer import Trainer
    torch.manual_seed(0)
    model = toy_model(20, 1)
    loader = prepare_dataloader(ToyDataset(128, seed=1), 32, shuffle=False)
    opt = FusedSGD(model.parameters(), lr=1e-3)
    tr = Trainer(model, loader, opt, "cpu", save_every=10**9, loss_fn="mse",
                 wrap_ddp=False, engine="auto",
                 checkpoint_path=str(tmp_path / "c.pt"))
    assert tr._engine is None  # fell back
    w0 = model.weight.detach().clone()
    tr.train(1)
    assert not torch.equal(model.weight.detach(), w0)


def test_reference_written_snapshot_loads(tmp_chdir):
    # a snapshot produced by the REFERENCE's literal code shape
    # (torch.nn.Linear state_dict under MODEL_STATE, multigpu_torchrun.py:57-62)
    # restores into this framework's Trainer/HipLinear unchanged
    import torch
    from mi355x_ddp.data import ToyDataset, prepare_dataloader
    from mi355x_ddp.models import toy_model
    from mi355x_ddp.trainer import Trainer

    ref_model = torch.nn.Linear(20, 1)
    torch.save({"MODEL_STATE": ref_model.state_dict(), "EPOCHS_RUN": 5},
               "snapshot.pt")

    model = toy_model(20, 1)
    loader = prepare_dataloader(ToyDataset(64, seed=0), 32, shuffle=False)
    opt = torch.optim.SGD(model.parameters(), lr=1e-3)
    tr = Trainer(model, loader, opt, "cpu", save_every=10**9, loss_fn="mse",
                 wrap_ddp=False, snapshot_path="snapshot.pt")
    assert tr.epochs_run == 5
    assert torch.equal(model.weight.detach(), ref_model.weight.detach())
    assert torch.equal(model.bias.detach(), ref_model.bias.detach())


def test_snapshot_save_is_atomic(tmp_chdir, monkeypatch):
    """A crash mid-save must leave the previous snapshot intact (the
    reference writes the live file in place — a mid-save kill corrupts
    snapshot.pt and turns every elastic restart into a crash loop)."""
    tr = _make(snapshot_path="snapshot.pt")
    tr._save_snapshot(3)
    good = torch.load("snapshot.pt", weights_only=True)
    assert good["EPOCHS_RUN"] == 3

    real_save = torch.save

    def crashing_save(obj, path):
        with open(path, "wb") as f:
            f.write(b"partial garbage")  # simulate dying mid-write
        raise RuntimeError("killed mid-save")

    monkeypatch.setattr(torch, "save", crashing_save)
    try:
        tr._save_snapshot(4)
    except RuntimeError:
        pass
    monkeypatch.setattr(torch, "save", real_save)

    # the live file still holds the epoch-3 snapshot, and no tmp debris
    again = torch.load("snapshot.pt", weights_only=True)
    assert again["EPOCHS_RUN"] == 3
    assert not [f for f in os.listdir(".") if ".tmp." in f]


def test_unknown_engine_rejected():
    import pytest
    ds = ToyDataset(64, seed=0)
    model = toy_model(20, 1)
    opt = FusedSGD(model.parameters(), lr=1e-3)
    dl = prepare_dataloader(ds, 32)
    with pytest.raises(ValueError, match="unknown engine"):
        Trainer(model, dl, opt, "cpu", 1, wrap_ddp=False, engine="presistent")


def test_framework_snapshot_loads_into_plain_nn_linear(tmp_chdir):
    """Reverse interop: a snapshot WE wrote restores into the reference's
    model type with the reference's loading pattern (torch.load +
    load_state_dict on nn.Linear(20,1); ref multigpu_torchrun.py:36-41) —
    the forward direction is test_reference_written_snapshot_loads."""
    tr = _make(snapshot_path="snapshot.pt")
    tr._save_snapshot(2)
    snap = torch.load("snapshot.pt", weights_only=True)
    ref = torch.nn.Linear(20, 1)
    ref.load_state_dict(snap["MODEL_STATE"])  # exact key/shape match
    assert snap["EPOCHS_RUN"] == 2
    assert torch.equal(ref.weight.data, tr._unwrapped().weight.data)


def test_ce_loss_class_index_targets_match_torch():
    """The common CrossEntropyLoss form (integer class indices, shape [B])
    works as a drop-in: same loss and same input gradients as torch."""
    from mi355x_ddp import ops

    torch.manual_seed(9)
    y = torch.randn(8, 5, requires_grad=True)
    idx = torch.randint(0, 5, (8,))
    loss = ops.cross_entropy(y, idx)
    want = torch.nn.functional.cross_entropy(y.detach(), idx)
    assert torch.allclose(loss, want, atol=1e-6)
    loss.backward()
    y2 = y.detach().clone().requires_grad_(True)
    torch.nn.functional.cross_entropy(y2, idx).backward()
    assert torch.allclose(y.grad, y2.grad, atol=1e-6)


def test_hooks_graph_engine_falls_back_on_cpu(tmp_path):
    # engine="hooks-graph" is a GPU optimization; on CPU the Trainer runs
    # the plain generic loop (no engine object) and trains identically
    import torch.distributed as dist

    from mi355x_ddp.data import MyTrainDataset, prepare_dataloader
    from mi355x_ddp.models import toy_model
    from mi355x_ddp.parallel import FusedSGD
    from mi355x_ddp.trainer import Trainer
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29787")
    dist.init_process_group("gloo", rank=0, world_size=1)
    try:
        torch.manual_seed(3)
        model = toy_model(20, 1)
        data = prepare_dataloader(MyTrainDataset(64), 32)
        opt = FusedSGD(model.parameters(), lr=1e-3)
        tr = Trainer(model, data, opt, gpu_id="cpu", save_every=1,
                     checkpoint_path=str(tmp_path / "c.pt"),
                     loss_fn="mse", engine="hooks-graph")
        assert tr._engine is None  # no graph engine off-GPU
        tr.train(2)
        assert (tmp_path / "c.pt").exists()
    finally:
        dist.destroy_process_group()
