"""Multi-process CPU tests of the native DDP engine over the gloo shadow
communicator (SURVEY §4 consequence (a)): the bucketing, hook, ordering and
averaging logic is identical to the GPU path; only the collective transport
differs.

Correctness oracle: DDP on 2 ranks, each seeing half of every batch, must
produce the same parameters as single-process training on the full batch
(mean losses make gradient averaging exact)."""

import os

import pytest
import torch
import torch.multiprocessing as mp

from mi355x_ddp import ops
from mi355x_ddp.models import toy_model
from mi355x_ddp.parallel import DDP, FusedSGD

WORLD = 2
STEPS = 5
LR = 0.05


def _make_model(seed):
    torch.manual_seed(seed)
    m = torch.nn.Sequential(
        toy_model(20, 16),
        torch.nn.ReLU(),
        toy_model(16, 1),
    )
    return m


def _reference_params(seed, data):
    model = _make_model(seed)
    opt = torch.optim.SGD(model.parameters(), lr=LR)
    for x, t in data:
        opt.zero_grad()
        loss = ops.mse_loss(model(x), t)
        loss.backward()
        opt.step()
    return [p.detach().clone() for p in model.parameters()]


def _free_port() -> int:
    import socket
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def _ddp_worker(rank, seed, data, port, grad_views, bucket_cap_mb, out_path):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    torch.distributed.init_process_group("gloo", rank=rank, world_size=WORLD)
    try:
        model = _make_model(seed)
        engine = DDP(model, bucket_cap_mb=bucket_cap_mb, grad_views=grad_views)
        opt = FusedSGD(model.parameters(), lr=LR)
        opt.attach_reducer(engine.reducer)
        half = 16
        for x, t in data:
            xs = x[rank * half:(rank + 1) * half]
            ts = t[rank * half:(rank + 1) * half]
            loss = ops.mse_loss(engine(xs), ts)
            loss.backward()
            engine.finalize_backward()
            opt.step()
        if rank == 0:
            torch.save([p.detach().clone() for p in model.parameters()],
                       out_path)
    finally:
        torch.distributed.destroy_process_group()


def _run_ddp(seed, data, port, grad_views, bucket_cap_mb=None, tmp_path="/tmp"):
    out_path = os.path.join(str(tmp_path), f"ddp_result_{port}.pt")
    if os.path.exists(out_path):
        os.remove(out_path)
    ctx = mp.get_context("spawn")
    procs = [ctx.Process(target=_ddp_worker,
                         args=(r, seed, data, port, grad_views,
                               bucket_cap_mb, out_path))
             for r in range(WORLD)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=180)
        assert p.exitcode == 0
    return torch.load(out_path, weights_only=True)


def _make_data(seed):
    g = torch.Generator().manual_seed(seed + 100)
    return [(torch.rand(32, 20, generator=g), torch.rand(32, 1, generator=g))
            for _ in range(STEPS)]


@pytest.mark.parametrize("grad_views", [True, False])
def test_ddp_matches_single_process(grad_views):
    seed = 1234
    data = _make_data(seed)
    ref = _reference_params(seed, data)
    got = _run_ddp(seed, data, port=_free_port(), grad_views=grad_views)
    for a, b in zip(got, ref):
        assert torch.allclose(a, b, atol=1e-6), (a - b).abs().max()


def test_ddp_many_buckets_ordering():
    # force one bucket per parameter (cap ~ 0) to exercise in-order launch
    seed = 77
    data = _make_data(seed)
    ref = _reference_params(seed, data)
    got = _run_ddp(seed, data, port=_free_port(), grad_views=True,
                   bucket_cap_mb=1e-6)
    for a, b in zip(got, ref):
        assert torch.allclose(a, b, atol=1e-6)


def test_reducer_bucket_assignment():
    from mi355x_ddp.parallel.reducer import Reducer
    model = _make_model(0)
    params = list(model.parameters())
    r = Reducer(params, comm=None, bucket_cap_mb=1e-6)
    # one bucket per param, reverse registration order
    assert len(r.buckets) == len(params)
    assert r.buckets[0].params[0] is params[-1]
    # params were rebound as views into flat buffers
    for b in r.buckets:
        for i, p in enumerate(b.params):
            assert p.data_ptr() == b.flat_param[b.offsets[i]:].data_ptr()
    # fused SGD on buckets == plain SGD
    model2 = _make_model(0)
    x = torch.rand(8, 20)
    t = torch.rand(8, 1)
    loss = ops.mse_loss(model(x), t)
    loss.backward()
    r.finalize()
    opt = FusedSGD(params, lr=0.1)
    opt.attach_reducer(r)
    opt.step()

    opt2 = torch.optim.SGD(model2.parameters(), lr=0.1)
    loss2 = ops.mse_loss(model2(x), t)
    loss2.backward()
    opt2.step()
    for a, b in zip(model.parameters(), model2.parameters()):
        assert torch.allclose(a, b, atol=1e-7)
    # grads were zeroed by the fused step
    for fp, fg in r.flat_pairs():
        assert fg.abs().sum() == 0


def test_ddp_setup_backend_defaults(monkeypatch):
    """GPU hosts must get the compound backend map: a bare "nccl" group
    rejects the CPU tensors that all_ranks_agree / mesh validation
    all-reduce over the default group (latent crash for every world>1
    entrypoint run on a real multi-GPU node)."""
    from mi355x_ddp.parallel import comm as C

    calls = {}
    monkeypatch.setenv("MASTER_ADDR", "127.0.0.1")
    monkeypatch.setenv("MASTER_PORT", "29999")
    monkeypatch.setattr(
        C.dist, "init_process_group",
        lambda backend, **kw: calls.__setitem__("backend", backend))

    monkeypatch.setattr(C.torch.cuda, "is_available", lambda: False)
    C.ddp_setup(0, 1)
    assert calls["backend"] == "gloo"

    monkeypatch.setattr(C.torch.cuda, "is_available", lambda: True)
    monkeypatch.setattr(C.torch.cuda, "set_device", lambda *_: None)
    C.ddp_setup(0, 1)
    assert calls["backend"] == "cpu:gloo,cuda:nccl"


def _accum_worker(rank, seed, data, port, grad_views, out_path):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    torch.distributed.init_process_group("gloo", rank=rank, world_size=WORLD)
    try:
        model = _make_model(seed)
        engine = DDP(model, grad_views=grad_views)
        opt = FusedSGD(model.parameters(), lr=LR)
        opt.attach_reducer(engine.reducer)
        half = 16
        # pairs of micro-batches: first under no_sync (local accumulation),
        # second with communication (reduces the accumulated totals)
        for (x1, t1), (x2, t2) in zip(data[0::2], data[1::2]):
            with engine.no_sync():
                loss = ops.mse_loss(
                    engine(x1[rank * half:(rank + 1) * half]),
                    t1[rank * half:(rank + 1) * half])
                loss.backward()
                engine.finalize_backward()
            loss = ops.mse_loss(
                engine(x2[rank * half:(rank + 1) * half]),
                t2[rank * half:(rank + 1) * half])
            loss.backward()
            engine.finalize_backward()
            opt.step()
        if rank == 0:
            torch.save([p.detach().clone() for p in model.parameters()],
                       out_path)
    finally:
        torch.distributed.destroy_process_group()


@pytest.mark.parametrize("grad_views", [True, False])
def test_no_sync_gradient_accumulation(grad_views, tmp_path):
    """DDP.no_sync accumulation == single-process training whose gradient
    is mean-over-shards of (g(micro1) + g(micro2))."""
    seed = 11
    data = _make_data(seed)  # 5 steps -> 2 accumulation pairs
    # single-process reference computing the identical update
    model = _make_model(seed)
    opt = torch.optim.SGD(model.parameters(), lr=LR)
    half = 16
    for (x1, t1), (x2, t2) in zip(data[0::2], data[1::2]):
        opt.zero_grad()
        acc = None
        for x, t in ((x1, t1), (x2, t2)):
            for r in range(WORLD):
                xs = x[r * half:(r + 1) * half]
                ts = t[r * half:(r + 1) * half]
                gs = torch.autograd.grad(
                    ops.mse_loss(model(xs), ts), list(model.parameters()))
                gs = [g / WORLD for g in gs]
                acc = gs if acc is None else [a + g for a, g in zip(acc, gs)]
        for p, g in zip(model.parameters(), acc):
            p.grad = g
        opt.step()
    want = [p.detach().clone() for p in model.parameters()]

    out_path = os.path.join(str(tmp_path), "accum.pt")
    ctx = mp.get_context("spawn")
    port = _free_port()
    procs = [ctx.Process(target=_accum_worker,
                         args=(r, seed, data, port, grad_views, out_path))
             for r in range(WORLD)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=180)
        assert p.exitcode == 0
    got = torch.load(out_path, weights_only=True)
    for g, w in zip(got, want):
        assert torch.allclose(g, w, atol=1e-6), (g - w).abs().max()


def _bufsync_worker(rank, port, out_path):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    torch.distributed.init_process_group("gloo", rank=rank, world_size=WORLD)
    try:
        torch.manual_seed(3)
        m = torch.nn.Sequential(torch.nn.BatchNorm1d(8), toy_model(8, 1))
        engine = DDP(m)
        bn = m[0]
        if rank == 1:  # drift rank 1's running stats
            with torch.no_grad():
                bn.running_mean.add_(5.0)
                bn.num_batches_tracked.add_(7)
        engine.eval()  # eval: forward does not update the stats itself
        engine(torch.rand(4, 8))
        if rank == 1:
            torch.save({"mean": bn.running_mean.clone(),
                        "nbt": bn.num_batches_tracked.clone()}, out_path)
    finally:
        torch.distributed.destroy_process_group()


def test_buffer_broadcast_every_forward(tmp_path):
    """stock-DDP broadcast_buffers parity: rank 0's buffers (fp32 running
    stats AND the int64 num_batches_tracked) overwrite drifted replicas
    before every forward."""
    out_path = os.path.join(str(tmp_path), "bufsync.pt")
    ctx = mp.get_context("spawn")
    port = _free_port()
    procs = [ctx.Process(target=_bufsync_worker, args=(r, port, out_path))
             for r in range(WORLD)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=180)
        assert p.exitcode == 0
    got = torch.load(out_path, weights_only=True)
    assert torch.all(got["mean"] == 0.0)  # rank 0's pristine running_mean
    assert int(got["nbt"]) == 0


def test_ddp_accepts_stock_signature():
    """The reference wraps with `DDP(model, device_ids=[gpu_id])`
    (ref multigpu.py:36); drop-in users keep that call form. On CPU,
    device_ids=["cpu"] validates; a device mismatch is loud."""
    m = toy_model(20, 1)
    DDP(m, device_ids=["cpu"])  # validates, no-op
    m2 = toy_model(20, 1)
    with pytest.raises(ValueError):
        DDP(m2, device_ids=[0])  # names cuda:0, model is on CPU


class _MaybeUnused(torch.nn.Module):
    def __init__(self):
        super().__init__()
        self.a = toy_model(20, 4)
        self.b = toy_model(4, 1)
        self.extra = toy_model(20, 1)  # unused on even steps
        self.use_extra = False

    def forward(self, x):
        y = self.b(torch.relu(self.a(x)))
        if self.use_extra:
            y = y + self.extra(x)
        return y


def _unused_worker(rank, port, out_path):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    torch.distributed.init_process_group("gloo", rank=rank, world_size=WORLD)
    try:
        torch.manual_seed(5)
        m = _MaybeUnused()
        engine = DDP(m)
        opt = FusedSGD(m.parameters(), lr=LR)
        opt.attach_reducer(engine.reducer)
        g = torch.Generator().manual_seed(50)
        for step in range(4):
            m.use_extra = step % 2 == 1  # extra's params unused half the time
            x = torch.rand(32, 20, generator=g)
            t = torch.rand(32, 1, generator=g)
            half = 16
            loss = ops.mse_loss(
                engine(x[rank * half:(rank + 1) * half]),
                t[rank * half:(rank + 1) * half])
            loss.backward()
            engine.finalize_backward()  # launches straggler buckets
            opt.step()
        if rank == 0:
            torch.save([p.detach().clone() for p in m.parameters()], out_path)
    finally:
        torch.distributed.destroy_process_group()


def test_unused_parameters_do_not_hang(tmp_path):
    """A model whose submodule is unused on some steps trains without
    find_unused_parameters-style flags: finalize() is an explicit
    all-ranks sync point, so straggler buckets launch in order with zero
    gradient segments (stock DDP HANGS here unless told in advance)."""
    out_path = os.path.join(str(tmp_path), "unused.pt")
    ctx = mp.get_context("spawn")
    port = _free_port()
    procs = [ctx.Process(target=_unused_worker, args=(r, port, out_path))
             for r in range(WORLD)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=180)
        assert p.exitcode == 0
    got = torch.load(out_path, weights_only=True)
    assert all(torch.isfinite(p).all() for p in got)


def _race_worker(rank, port, out_dir):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["MI355X_DEBUG_SYNC"] = "1"
    torch.distributed.init_process_group("gloo", rank=rank, world_size=WORLD)
    try:
        model = _make_model(0)
        engine = DDP(model)
        opt = FusedSGD(model.parameters(), lr=LR)
        opt.attach_reducer(engine.reducer)
        x, t = torch.randn(8, 20), torch.randn(8, 1)
        loss = ops.mse_loss(engine(x), t)
        loss.backward()
        # DELIBERATE ordering corruption: step() without finalize —
        # collectives launched by the hooks are still unfenced
        caught = False
        try:
            opt.step()
        except RuntimeError as e:
            caught = "unfenced" in str(e)
        # recover properly and verify the good ordering passes
        engine.finalize_backward()
        opt.step()
        with open(os.path.join(out_dir, f"race{rank}.ok"), "w") as f:
            f.write("caught" if caught else "missed")
        torch.distributed.barrier()
    finally:
        torch.distributed.destroy_process_group()


def test_debug_sync_catches_step_before_finalize(tmp_path):
    """the reducer's one real race is reading flat_grad while
    bucket collectives are unfenced. MI355X_DEBUG_SYNC=1 must catch the
    corrupted ordering (step before finalize) and pass the correct one."""
    mp.spawn(_race_worker, args=(_free_port(), str(tmp_path)),
             nprocs=WORLD, join=True)
    for r in range(WORLD):
        assert (tmp_path / f"race{r}.ok").read_text() == "caught"


def test_ddp_accepts_stock_kwargs():
    # full drop-in surface: stock DDP's remaining kwargs are accepted
    # (find_unused_parameters/static_graph no-ops — unused params train
    # by default here; gradient_as_bucket_view maps to grad_views)
    m = _make_model(1)
    eng = DDP(m, find_unused_parameters=True, static_graph=True,
              gradient_as_bucket_view=True, broadcast_buffers=True)
    assert eng.reducer.grad_views is True
    m2 = _make_model(1)
    eng2 = DDP(m2, gradient_as_bucket_view=False)
    assert eng2.reducer.grad_views is False
