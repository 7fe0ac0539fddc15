"""GPU tests for the device-side xGMI mesh all-reduce (P2pMesh): several
processes on ONE device exchange IPC mailboxes — same protocol the 8-GPU
node runs over xGMI. Verifies the raw mesh, the adapter (incl. its gloo
cross-validation), and a world-2 fused-engine training run against the
pure-gloo transport."""

import os

import pytest
import torch
import torch.multiprocessing as mp

pytestmark = pytest.mark.gpu


def _free_port():
    import socket
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def _init(rank, world, port):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    torch.distributed.init_process_group("gloo", rank=rank, world_size=world)
    torch.cuda.set_device(0)


def _mesh_worker(rank, world, port, out):
    from mi355x_ddp import ops
    _init(rank, world, port)
    try:
        from mi355x_ddp.parallel.comm import GlooComm, P2pMeshComm
        comm = P2pMeshComm(torch.device("cuda", 0), base=GlooComm())
        t = (torch.arange(24, dtype=torch.float32, device="cuda") + 1) \
            * (rank + 1)
        expect = t.cpu().clone()
        torch.distributed.all_reduce(expect)
        expect /= world
        comm.all_reduce_avg_inline(t)
        torch.cuda.synchronize()
        comm.check()
        assert torch.allclose(t.cpu(), expect, atol=1e-6), \
            (t.cpu() - expect).abs().max()
        # bf16 payload too
        tb = (torch.arange(22, device="cuda").bfloat16() + 1) * (rank + 1)
        eb = tb.float().cpu()
        torch.distributed.all_reduce(eb)
        eb /= world
        comm.all_reduce_avg_inline(tb)
        torch.cuda.synchronize()
        comm.check()
        assert torch.allclose(tb.float().cpu(), eb, atol=0.25), \
            (tb.float().cpu() - eb).abs().max()
        if rank == 0:
            torch.save({"ok": True}, out)
    finally:
        torch.distributed.destroy_process_group()


@pytest.mark.parametrize("world", [2, 4])
def test_mesh_allreduce_multiprocess_one_device(world, tmp_path):
    out = str(tmp_path / "ok.pt")
    mp.spawn(_mesh_worker, args=(world, _free_port(), out), nprocs=world,
             join=True)
    assert torch.load(out, weights_only=True)["ok"]


def _train_worker(rank, world, port, use_mesh, out_dir):
    from mi355x_ddp.engine import ToyFusedStep
    from mi355x_ddp.models import toy_model
    from mi355x_ddp.parallel.comm import GlooComm, P2pMeshComm
    _init(rank, world, port)
    try:
        comm = GlooComm()
        if use_mesh:
            comm = P2pMeshComm(torch.device("cuda", 0), base=comm)
        torch.manual_seed(11)
        model = toy_model(20, 1).to("cuda")
        eng = ToyFusedStep(model, comm=comm, lr=0.05, use_mse=True)
        eng.reducer.broadcast_params(root=0)
        g = torch.Generator().manual_seed(100 + rank)
        X = torch.rand(12, 32, 20, generator=g).to("cuda")
        T = torch.rand(12, 32, 1, generator=g).to("cuda")
        for s in range(12):
            eng.step(X[s], T[s])
        torch.cuda.synchronize()
        if hasattr(comm, "check"):
            comm.check()
        if rank == 0:
            torch.save({"w": model.weight.detach().cpu(),
                        "b": model.bias.detach().cpu()},
                       os.path.join(out_dir, f"mesh{int(use_mesh)}.pt"))
        torch.distributed.barrier()
    finally:
        torch.distributed.destroy_process_group()


def test_fused_engine_world2_mesh_matches_gloo(tmp_path):
    for use_mesh in (False, True):
        mp.spawn(_train_worker,
                 args=(2, _free_port(), use_mesh, str(tmp_path)), nprocs=2,
                 join=True)
    a = torch.load(tmp_path / "mesh0.pt", weights_only=True)
    b = torch.load(tmp_path / "mesh1.pt", weights_only=True)
    assert torch.allclose(a["w"], b["w"], atol=1e-6), \
        (a["w"] - b["w"]).abs().max()
    assert torch.allclose(a["b"], b["b"], atol=1e-6)


def _persistent_worker(rank, world, port, out_dir):
    from mi355x_ddp.engine import PersistentToyStep
    from mi355x_ddp.models import toy_model
    from mi355x_ddp.parallel.comm import GlooComm, P2pMeshComm
    _init(rank, world, port)
    try:
        comm = P2pMeshComm(torch.device("cuda", 0), base=GlooComm())
        torch.manual_seed(11)
        model = toy_model(20, 1).to("cuda")
        eng = PersistentToyStep(model, comm=comm, lr=0.05, use_mse=True)
        eng.reducer.broadcast_params(root=0)
        g = torch.Generator().manual_seed(100 + rank)
        Xf = torch.rand(12 * 32, 20, generator=g).to("cuda")
        Tf = torch.rand(12 * 32, 1, generator=g).to("cuda")
        eng.bind_shard(Xf, Tf, 32)
        for s in range(12):
            eng.step_shard(s)
        eng.flush()
        torch.cuda.synchronize()
        comm.check()
        if rank == 0:
            torch.save({"w": model.weight.detach().cpu(),
                        "b": model.bias.detach().cpu()},
                       os.path.join(out_dir, "persistent_mesh.pt"))
        torch.distributed.barrier()
    finally:
        torch.distributed.destroy_process_group()


def test_persistent_mesh_world2_matches_fused_gloo(tmp_path):
    # same data/ordering as _train_worker's gloo run: deferred multi-step
    # kernel with an IN-KERNEL mesh all-reduce per step == per-step fused
    # kernel + gloo all-reduce
    mp.spawn(_train_worker, args=(2, _free_port(), False, str(tmp_path)),
             nprocs=2, join=True)
    mp.spawn(_persistent_worker, args=(2, _free_port(), str(tmp_path)),
             nprocs=2, join=True)
    a = torch.load(tmp_path / "mesh0.pt", weights_only=True)
    b = torch.load(tmp_path / "persistent_mesh.pt", weights_only=True)
    assert torch.allclose(a["w"], b["w"], atol=1e-5), \
        (a["w"] - b["w"]).abs().max()
    assert torch.allclose(a["b"], b["b"], atol=1e-5)


def _persistent_bf16_worker(rank, world, port, out_dir):
    from mi355x_ddp.engine import PersistentToyStep
    from mi355x_ddp.models import toy_model
    from mi355x_ddp.parallel.comm import GlooComm, P2pMeshComm
    _init(rank, world, port)
    try:
        comm = P2pMeshComm(torch.device("cuda", 0), base=GlooComm())
        torch.manual_seed(11)
        model = toy_model(20, 1).to("cuda").bfloat16()
        eng = PersistentToyStep(model, comm=comm, lr=0.05, use_mse=True)
        eng.reducer.broadcast_params(root=0)
        g = torch.Generator().manual_seed(100 + rank)
        Xf = torch.rand(10 * 32, 20, generator=g).to("cuda").bfloat16()
        Tf = torch.rand(10 * 32, 1, generator=g).to("cuda").bfloat16()
        eng.bind_shard(Xf, Tf, 32)
        for s in range(10):
            eng.step_shard(s)
        eng.flush()
        torch.cuda.synchronize()
        comm.check()
        w = model.weight.detach().float().cpu()
        assert torch.isfinite(w).all()
        if rank == 0:
            torch.save({"w": w}, os.path.join(out_dir, "bf16.pt"))
        torch.distributed.barrier()
    finally:
        torch.distributed.destroy_process_group()


def test_persistent_mesh_bf16_world2(tmp_path):
    mp.spawn(_persistent_bf16_worker, args=(2, _free_port(), str(tmp_path)),
             nprocs=2, join=True)
    w = torch.load(tmp_path / "bf16.pt", weights_only=True)["w"]
    assert torch.isfinite(w).all() and w.abs().sum() > 0


def _timeout_worker(rank, world, port, out):
    from mi355x_ddp import ops
    _init(rank, world, port)
    try:
        from mi355x_ddp.parallel.comm import GlooComm, P2pMeshComm
        comm = P2pMeshComm(torch.device("cuda", 0), base=GlooComm())
        if rank == 0:
            # rank 1 never issues this exchange: the bounded spin must
            # expire and check() must raise — never a hang
            t = torch.ones(8, device="cuda")
            comm._mesh.all_reduce_avg_inline(t)
            torch.cuda.synchronize()
            err = None
            try:
                comm.check()
            except RuntimeError as e:
                err = str(e)
            assert err is not None and "timed out" in err
            torch.save({"raised": True}, out)
        # rank 1 just waits so rank 0's kernel canobserve its absence, then
        # both leave together
        torch.distributed.barrier()
    finally:
        torch.distributed.destroy_process_group()


def test_mesh_timeout_raises_not_hangs(tmp_path):
    out = str(tmp_path / "t.pt")
    mp.spawn(_timeout_worker, args=(2, _free_port(), out), nprocs=2,
             join=True)
    assert torch.load(out, weights_only=True)["raised"]


def _trainer_mesh_worker(rank, world, port, out_dir):
    from mi355x_ddp.data import ToyDataset, prepare_dataloader
    from mi355x_ddp.models import toy_model
    from mi355x_ddp.parallel import FusedSGD
    from mi355x_ddp.trainer import Trainer
    _init(rank, world, port)
    try:
        torch.manual_seed(2)
        model = toy_model(20, 1)
        loader = prepare_dataloader(ToyDataset(256, seed=9), 32,
                                    distributed=True, shuffle=False,
                                    num_replicas=world, rank=rank)
        opt = FusedSGD(model.parameters(), lr=0.05)
        tr = Trainer(model, loader, opt, 0, save_every=10**9, loss_fn="mse",
                     engine="persistent",
                     checkpoint_path=os.path.join(out_dir, "c.pt"))
        assert tr._engine is not None
        assert getattr(tr._engine, "_mesh", None) is not None
        tr.train(2)
        torch.cuda.synchronize()
        if rank == 0:
            torch.save({"w": model.weight.detach().cpu()},
                       os.path.join(out_dir, "trainer_mesh.pt"))
        torch.distributed.barrier()
    finally:
        torch.distributed.destroy_process_group()


def test_trainer_persistent_mesh_world2(tmp_path):
    # the public Trainer API drives the multi-step + in-kernel-mesh engine
    # at world 2 (one device)
    mp.spawn(_trainer_mesh_worker, args=(2, _free_port(), str(tmp_path)),
             nprocs=2, join=True)
    w = torch.load(tmp_path / "trainer_mesh.pt", weights_only=True)["w"]
    assert torch.isfinite(w).all() and w.abs().sum() > 0


def _persistent_b64_worker(rank, world, port, dtype_name, out_dir):
    # batch-64 instantiations of the MESH multistep kernels (r02: the
    # fast-path matrix covers B in {32,64} x {f32,bf16} at any world)
    from mi355x_ddp.engine import PersistentToyStep
    from mi355x_ddp.models import toy_model
    from mi355x_ddp.parallel.comm import GlooComm, P2pMeshComm
    _init(rank, world, port)
    dtype = torch.bfloat16 if dtype_name == "bf16" else torch.float32
    try:
        comm = P2pMeshComm(torch.device("cuda", 0), base=GlooComm())
        torch.manual_seed(13)
        model = toy_model(20, 1).to("cuda").to(dtype)
        eng = PersistentToyStep(model, comm=comm, lr=0.04, use_mse=True)
        eng.reducer.broadcast_params(root=0)
        g = torch.Generator().manual_seed(200 + rank)
        Xf = torch.rand(6 * 64, 20, generator=g).to("cuda").to(dtype)
        Tf = torch.rand(6 * 64, 1, generator=g).to("cuda").to(dtype)
        eng.bind_shard(Xf, Tf, 64)
        for s in range(6):
            eng.step_shard(s)
        eng.flush()
        torch.cuda.synchronize()
        comm.check()
        if rank == 0:
            torch.save({"w": model.weight.detach().float().cpu(),
                        "b": model.bias.detach().float().cpu()},
                       os.path.join(out_dir, f"pm64_{dtype_name}.pt"))
        torch.distributed.barrier()
    finally:
        torch.distributed.destroy_process_group()


def _fused_b64_worker(rank, world, port, dtype_name, out_dir):
    from mi355x_ddp.engine import ToyFusedStep
    from mi355x_ddp.models import toy_model
    from mi355x_ddp.parallel.comm import GlooComm
    _init(rank, world, port)
    dtype = torch.bfloat16 if dtype_name == "bf16" else torch.float32
    try:
        comm = GlooComm()
        torch.manual_seed(13)
        model = toy_model(20, 1).to("cuda").to(dtype)
        eng = ToyFusedStep(model, comm=comm, lr=0.04, use_mse=True)
        eng.reducer.broadcast_params(root=0)
        g = torch.Generator().manual_seed(200 + rank)
        Xf = torch.rand(6 * 64, 20, generator=g).to("cuda").to(dtype)
        Tf = torch.rand(6 * 64, 1, generator=g).to("cuda").to(dtype)
        for s in range(6):
            eng.step(Xf[s * 64:(s + 1) * 64].contiguous(),
                     Tf[s * 64:(s + 1) * 64].contiguous())
        torch.cuda.synchronize()
        if rank == 0:
            torch.save({"w": model.weight.detach().float().cpu(),
                        "b": model.bias.detach().float().cpu()},
                       os.path.join(out_dir, f"fused64_{dtype_name}.pt"))
        torch.distributed.barrier()
    finally:
        torch.distributed.destroy_process_group()


@pytest.mark.parametrize("dtype_name", ["fp32", "bf16"])
def test_persistent_mesh_world2_batch64(dtype_name, tmp_path):
    mp.spawn(_fused_b64_worker,
             args=(2, _free_port(), dtype_name, str(tmp_path)),
             nprocs=2, join=True)
    mp.spawn(_persistent_b64_worker,
             args=(2, _free_port(), dtype_name, str(tmp_path)),
             nprocs=2, join=True)
    a = torch.load(tmp_path / f"fused64_{dtype_name}.pt", weights_only=True)
    b = torch.load(tmp_path / f"pm64_{dtype_name}.pt", weights_only=True)
    tol = 5e-3 if dtype_name == "bf16" else 1e-5
    assert torch.allclose(a["w"], b["w"], atol=tol, rtol=5e-2), \
        (a["w"] - b["w"]).abs().max()
    assert torch.allclose(a["b"], b["b"], atol=tol, rtol=5e-2)
