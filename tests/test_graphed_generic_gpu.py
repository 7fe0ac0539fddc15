"""GPU tests for GraphedAutogradStep — whole-step hipGraph capture of the
GENERIC autograd path (fwd + loss + backward + ReducerCore hooks + fused
SGD in one replayed graph). Oracle: the same path run eager, with the
capture call's warmup accounting replicated, so parity is bitwise."""

import os

import pytest
import torch
from torch import nn

from mi355x_ddp import ops
from mi355x_ddp.engine import GraphedAutogradStep
from mi355x_ddp.parallel import DDP, FusedSGD

pytestmark = pytest.mark.gpu

DEV = "cuda:0"


def _mlp(seed):
    torch.manual_seed(seed)
    return nn.Sequential(
        nn.Linear(64, 256), nn.ReLU(),
        nn.Linear(256, 256), nn.ReLU(),
        nn.Linear(256, 10)).to(DEV)


def _batches(n, seed=21):
    g = torch.Generator().manual_seed(seed)
    return [(torch.randn(16, 64, generator=g).to(DEV),
             torch.randn(16, 10, generator=g).to(DEV)) for _ in range(n)]


def _build(seed):
    # capture needs Python hooks (C++ node post-hooks segfault
    # hipStreamEndCapture on this build — reducer.py); replay runs none
    m = _mlp(seed)
    eng = DDP(m, comm=None, cpp_hooks=False)
    opt = FusedSGD(m.parameters(), lr=0.01)
    opt.attach_reducer(eng.reducer)
    return m, eng, opt


def test_graphed_step_matches_eager_bitwise():
    W = 3  # GraphedAutogradStep default warmup_steps
    data = _batches(6)

    m_g, eng_g, opt_g = _build(0)
    gs = GraphedAutogradStep(eng_g, ops.mse_loss, opt_g,
                             finalize=eng_g.finalize_backward,
                             warmup_steps=W)
    for x, t in data:
        gs.step(x, t)
    torch.cuda.synchronize()
    assert not gs._broken and 1 in gs._graphs, "capture must succeed for MLP"

    # eager arm replicating the capture call's schedule: W warmup steps on
    # batch0, then the captured step on batch0, then batches 1..5
    m_e, eng_e, opt_e = _build(0)

    def eager(x, t):
        loss = ops.mse_loss(eng_e(x), t)
        loss.backward()
        eng_e.finalize_backward()
        opt_e.step()

    for _ in range(W + 1):
        eager(*data[0])
    for x, t in data[1:]:
        eager(x, t)
    torch.cuda.synchronize()
    for a, b in zip(m_g.parameters(), m_e.parameters()):
        assert torch.equal(a, b), (a - b).abs().max()


def test_graphed_step_with_world1_rccl_capture():
    # capture must record the C++ hook's ncclAllReduce + join fence; at
    # world 1 the average is the identity, so parity vs comm-less holds
    import torch.distributed as dist
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29782")
    dist.init_process_group("gloo", rank=0, world_size=1)
    try:
        from mi355x_ddp.parallel.comm import RcclCommAdapter
        comm = RcclCommAdapter(torch.device(DEV))
        data = _batches(5, seed=33)

        m_g = _mlp(1)
        eng_g = DDP(m_g, comm=comm, bucket_cap_mb=0.05, cpp_hooks=False)
        assert eng_g.reducer._core is None  # Python hooks: capture-safe
        opt_g = FusedSGD(m_g.parameters(), lr=0.01)
        opt_g.attach_reducer(eng_g.reducer)
        gs = GraphedAutogradStep(eng_g, ops.mse_loss, opt_g,
                                 finalize=eng_g.finalize_backward)
        for x, t in data:
            gs.step(x, t)
        torch.cuda.synchronize()
        assert not gs._broken and 1 in gs._graphs, \
            "capture with an in-graph RCCL collective must succeed"

        m_e = _mlp(1)
        eng_e = DDP(m_e, comm=None, bucket_cap_mb=0.05, cpp_hooks=False)
        opt_e = FusedSGD(m_e.parameters(), lr=0.01)
        opt_e.attach_reducer(eng_e.reducer)
        ge = GraphedAutogradStep(eng_e, ops.mse_loss, opt_e,
                                 finalize=eng_e.finalize_backward)
        for x, t in data:
            ge.step(x, t)
        torch.cuda.synchronize()
        for a, b in zip(m_g.parameters(), m_e.parameters()):
            assert torch.allclose(a, b, atol=1e-6), (a - b).abs().max()
    finally:
        dist.destroy_process_group()


def test_graphed_resnet50_steps():
    # MIOpen conv/BN under whole-step capture: allowed to fall back to
    # eager (with a warning) — training must be correct either way
    from mi355x_ddp.models import resnet50
    torch.manual_seed(0)
    m = resnet50().to(DEV)
    eng = DDP(m, comm=None, cpp_hooks=False)
    opt = FusedSGD(m.parameters(), lr=1e-4)
    opt.attach_reducer(eng.reducer)
    gs = GraphedAutogradStep(eng, ops.cross_entropy, opt, warmup_steps=2,
                             finalize=eng.finalize_backward)
    x = torch.rand(8, 3, 224, 224, device=DEV)
    t = torch.rand(8, 1000, device=DEV)
    import warnings
    with warnings.catch_warnings():
        warnings.simplefilter("ignore", RuntimeWarning)
        for _ in range(3):
            gs.step(x, t)
    torch.cuda.synchronize()
    assert all(torch.isfinite(p).all() for p in m.parameters())


def test_trainer_hooks_graph_engine(tmp_path):
    # Trainer(engine="hooks-graph") trains the toy through the captured
    # generic path; weights match the eager hooks engine bitwise
    import torch.distributed as dist

    from mi355x_ddp.data import MyTrainDataset, prepare_dataloader
    from mi355x_ddp.models import toy_model
    from mi355x_ddp.trainer import Trainer

    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29783")
    dist.init_process_group("gloo", rank=0, world_size=1)
    try:
        torch.manual_seed(77)
        model = toy_model(20, 1).to(DEV)
        init = [p.detach().clone() for p in model.parameters()]
        data = prepare_dataloader(MyTrainDataset(128), 32)
        opt = FusedSGD(model.parameters(), lr=1e-3)
        ck = str(tmp_path / "hg.pt")
        tr = Trainer(model, data, opt, gpu_id=0, save_every=1,
                     checkpoint_path=ck, loss_fn="mse",
                     engine="hooks-graph")
        assert tr._engine is not None, "hooks-graph must install the engine"
        tr.train(3)
        torch.cuda.synchronize()
        assert not tr._engine._broken and tr._engine._graphs, \
            "toy step must capture"
        got = [p.detach() for p in tr._unwrapped().parameters()]
        assert all(torch.isfinite(p).all() for p in got)
        assert any(not torch.equal(a, b) for a, b in zip(init, got)), \
            "training must have updated the weights"
        # checkpoint written and loadable into a plain module
        sd = torch.load(ck, map_location="cpu", weights_only=True)
        toy_model(20, 1).load_state_dict(sd)
    finally:
        dist.destroy_process_group()


def test_graphed_shard_bound_matches_eager_bitwise():
    # the multi-step chunked-replay path (bind_shard/step_shard/flush):
    # S=13 decomposes as one 8-step graph + five 1-step graphs
    S, B = 13, 16
    g = torch.Generator().manual_seed(55)
    xs = torch.randn(S * B, 64, generator=g).to(DEV)
    ts = torch.randn(S * B, 10, generator=g).to(DEV)

    m_g, eng_g, opt_g = _build(6)
    gs = GraphedAutogradStep(eng_g, ops.mse_loss, opt_g,
                             finalize=eng_g.finalize_backward,
                             chunk_sizes=(8, 1))
    gs.bind_shard(xs, ts, B)
    for i in range(S):
        gs.step_shard(i)
    gs.flush()
    torch.cuda.synchronize()
    assert not gs._broken and 8 in gs._graphs and 1 in gs._graphs

    # eager arm: warmup trains warmup_steps times on batch 0, then each
    # batch in order
    m_e, eng_e, opt_e = _build(6)

    def eager(lo, hi):
        loss = ops.mse_loss(eng_e(xs[lo:hi]), ts[lo:hi])
        loss.backward()
        eng_e.finalize_backward()
        opt_e.step()

    for _ in range(gs.warmup_steps):
        eager(0, B)
    for i in range(S):
        eager(i * B, (i + 1) * B)
    torch.cuda.synchronize()
    for a, b in zip(m_g.parameters(), m_e.parameters()):
        assert torch.equal(a, b), (a - b).abs().max()


def test_graphed_toy_step_with_world1_rccl():
    # GraphedToyStep capture under a communicator was
    # untested. At world 1 the RCCL inline all-reduce is the identity, so
    # the captured toy step must match the comm-less fused step bitwise.
    import torch.distributed as dist

    from mi355x_ddp.engine import GraphedToyStep, ToyFusedStep
    from mi355x_ddp.models import toy_model
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29785")
    dist.init_process_group("gloo", rank=0, world_size=1)
    try:
        from mi355x_ddp.parallel.comm import RcclCommAdapter
        comm = RcclCommAdapter(torch.device(DEV))
        g = torch.Generator().manual_seed(41)
        X = torch.rand(8, 32, 20, generator=g).to(DEV)
        T = torch.rand(8, 32, 1, generator=g).to(DEV)

        torch.manual_seed(9)
        m_g = toy_model(20, 1).to(DEV)
        eng_g = GraphedToyStep(m_g, comm=comm, lr=0.03, use_mse=True)
        for s in range(8):
            eng_g.step(X[s].contiguous(), T[s].contiguous())
        torch.cuda.synchronize()
        assert eng_g._graph not in (None, False), \
            "capture with the inline RCCL collective must succeed"

        torch.manual_seed(9)
        m_e = toy_model(20, 1).to(DEV)
        eng_e = ToyFusedStep(m_e, comm=None, lr=0.03, use_mse=True)
        # replicate the capture call's accounting: call 0 runs one eager
        # step (warmup) and the captured step is NOT executed that call
        eng_e.step(X[0].contiguous(), T[0].contiguous())
        for s in range(1, 8):
            eng_e.step(X[s].contiguous(), T[s].contiguous())
        torch.cuda.synchronize()
        assert torch.allclose(m_g.weight.detach(), m_e.weight.detach(),
                              atol=1e-6), \
            (m_g.weight - m_e.weight).abs().max()
    finally:
        dist.destroy_process_group()


def test_graphed_shard_rebind_with_new_batch_size():
    # binding a shard with a DIFFERENT batch size drops the captured
    # graphs and recaptures at the new shape (stale fixed-shape buffers
    # would otherwise corrupt the copy); training stays correct
    m, eng, opt = _build(7)
    gs = GraphedAutogradStep(eng, ops.mse_loss, opt,
                             finalize=eng.finalize_backward,
                             chunk_sizes=(4, 1))
    g1 = torch.Generator().manual_seed(71)
    xs1 = torch.randn(4 * 16, 64, generator=g1).to(DEV)
    ts1 = torch.randn(4 * 16, 10, generator=g1).to(DEV)
    gs.bind_shard(xs1, ts1, 16)
    for i in range(4):
        gs.step_shard(i)
    gs.flush()
    xs2 = torch.randn(4 * 8, 64, generator=g1).to(DEV)
    ts2 = torch.randn(4 * 8, 10, generator=g1).to(DEV)
    gs.bind_shard(xs2, ts2, 8)  # batch 16 -> 8: graphs invalidated
    for i in range(4):
        gs.step_shard(i)
    gs.flush()
    torch.cuda.synchronize()
    assert not gs._broken and 4 in gs._graphs

    m_e, eng_e, opt_e = _build(7)

    def eager(x, t):
        loss = ops.mse_loss(eng_e(x), t)
        loss.backward()
        eng_e.finalize_backward()
        opt_e.step()

    for _ in range(gs.warmup_steps):
        eager(xs1[:16], ts1[:16])
    for i in range(4):
        eager(xs1[i * 16:(i + 1) * 16], ts1[i * 16:(i + 1) * 16])
    for i in range(4):
        eager(xs2[i * 8:(i + 1) * 8], ts2[i * 8:(i + 1) * 8])
    torch.cuda.synchronize()
    for a, b in zip(m.parameters(), m_e.parameters()):
        assert torch.equal(a, b), (a - b).abs().max()


def test_graphed_state_machine_fuzz():
    """Randomized interleaving of the engine's surfaces — sequential
    step_shard runs, run-breaking random indices, tensor-API step()
    calls, explicit flushes, shard re-binds — against an eager oracle
    that replays the exact submission order (mirror of the persistent
    engine's fuzz, test_engine_gpu.py)."""
    import random
    rng = random.Random(1234)
    S, B = 24, 16
    g = torch.Generator().manual_seed(99)
    xs = torch.randn(S * B, 64, generator=g).to(DEV)
    ts = torch.randn(S * B, 10, generator=g).to(DEV)
    pool_x = torch.randn(8 * B, 64, generator=g).to(DEV)
    pool_t = torch.randn(8 * B, 10, generator=g).to(DEV)

    for trial in range(3):
        m, eng, opt = _build(100 + trial)
        gs = GraphedAutogradStep(eng, ops.mse_loss, opt,
                                 finalize=eng.finalize_backward,
                                 chunk_sizes=(8, 1))
        m_e, eng_e, opt_e = _build(100 + trial)

        def eager(x, t):
            loss = ops.mse_loss(eng_e(x), t)
            loss.backward()
            eng_e.finalize_backward()
            opt_e.step()

        submitted = []
        gs.bind_shard(xs, ts, B)
        next_seq = 0
        warm_done = False
        for op in range(40):
            r = rng.random()
            if r < 0.45:  # sequential shard index
                i = next_seq % S
                next_seq += 1
                gs.step_shard(i)
                submitted.append((xs[i * B:(i + 1) * B],
                                  ts[i * B:(i + 1) * B]))
            elif r < 0.65:  # random index (breaks the run)
                i = rng.randrange(S)
                next_seq = i + 1
                gs.step_shard(i)
                submitted.append((xs[i * B:(i + 1) * B],
                                  ts[i * B:(i + 1) * B]))
            elif r < 0.85:  # tensor-API step from a different buffer
                j = rng.randrange(8)
                x, t = (pool_x[j * B:(j + 1) * B],
                        pool_t[j * B:(j + 1) * B])
                gs.step(x, t)
                submitted.append((x, t))
            else:
                gs.flush()
            # the first executed work triggers the one-time warmup (3
            # extra steps on that batch) — replicate in the oracle
            if submitted and not warm_done:
                # warmup happens when the first step/flush actually runs
                if gs._warmed:
                    warm_done = True
                    x0, t0 = submitted[0]
                    for _ in range(gs.warmup_steps):
                        eager(x0, t0)
        gs.flush()
        torch.cuda.synchronize()
        assert not gs._broken
        if not warm_done and submitted and gs._warmed:
            x0, t0 = submitted[0]
            for _ in range(gs.warmup_steps):
                eager(x0, t0)
        for x, t in submitted:
            eager(x, t)
        torch.cuda.synchronize()
        for a, b in zip(m.parameters(), m_e.parameters()):
            assert torch.equal(a, b), (trial, (a - b).abs().max())


def test_graphed_recovers_from_invalidated_capture():
    # force a capture invalidation deterministically (a synchronize mid-
    # capture is illegal) and verify the engine recovers: warning, eager
    # fallback, training continues and stays correct (engine.py
    # _recover_failed_capture — exercised for real by MIOpen NHWC
    # workspace allocs, profiles r02d)
    import warnings as warnings_mod

    m, eng, opt = _build(8)
    state = {"broke": False}

    def poisoned_loss(y, t):
        if torch.cuda.is_current_stream_capturing() and not state["broke"]:
            state["broke"] = True
            torch.cuda.synchronize()  # invalidates the capture
        return ops.mse_loss(y, t)

    gs = GraphedAutogradStep(eng, poisoned_loss, opt,
                             finalize=eng.finalize_backward)
    data = _batches(4, seed=91)
    with warnings_mod.catch_warnings(record=True) as w:
        warnings_mod.simplefilter("always")
        for x, t in data:
            gs.step(x, t)
    torch.cuda.synchronize()
    assert gs._broken
    assert any("capture failed" in str(x.message) for x in w)
    # the default CUDA generator must be usable again (a failed capture
    # leaves its capture flag stuck without the recovery's state swap)
    assert torch.isfinite(torch.randn(8, device=DEV)).all()
    # process-wide poison: a NEW engine must not attempt to capture (a
    # second capture_begin after an invalidated capture is lethal on this
    # build — engine.py _poison_captures; tools/recapture_probe.py).
    # This test therefore runs LAST in the file.
    import mi355x_ddp.engine as eng_mod
    assert eng_mod._CAPTURE_POISONED
    m2, eng2, opt2 = _build(9)
    gs2 = GraphedAutogradStep(eng2, ops.mse_loss, opt2,
                              finalize=eng2.finalize_backward)
    for x, t in _batches(2, seed=92):
        gs2.step(x, t)
    torch.cuda.synchronize()
    assert gs2._broken and not gs2._graphs  # eager, no capture attempted
    assert all(torch.isfinite(p).all() for p in m2.parameters())

    # eager arm: identical schedule (warmup trains, the poisoned capture
    # executed nothing, every step() ran eager)
    m_e, eng_e, opt_e = _build(8)

    def eager(x, t):
        loss = ops.mse_loss(eng_e(x), t)
        loss.backward()
        eng_e.finalize_backward()
        opt_e.step()

    for _ in range(gs.warmup_steps):
        eager(*data[0])
    for x, t in data:
        eager(x, t)
    torch.cuda.synchronize()
    for a, b in zip(m.parameters(), m_e.parameters()):
        assert torch.equal(a, b), (a - b).abs().max()
