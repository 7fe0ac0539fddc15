"""CPU property tests: Reducer bucket-assignment invariants over random
parameter lists, and the hang-safety agreement logic of the transport
ladder under PARTIAL per-rank failure."""

import os

import torch
from torch import nn


def test_reducer_bucket_invariants_fuzz():
    from hypothesis import given, settings, strategies as st
    from mi355x_ddp.parallel.reducer import Reducer

    @settings(max_examples=50, deadline=None)
    @given(shapes=st.lists(st.tuples(st.integers(1, 40), st.integers(1, 40)),
                           min_size=1, max_size=12),
           cap_kb=st.sampled_from([0.25, 1.0, 16.0, 1024.0]),
           views=st.booleans())
    def check(shapes, cap_kb, views):
        params = [nn.Parameter(torch.randn(*s)) for s in shapes]
        originals = [p.detach().clone() for p in params]
        red = Reducer(params, comm=None, bucket_cap_mb=cap_kb / 1024,
                      grad_views=views)
        seen = set()
        order = []
        for b in red.buckets:
            assert b.flat_param.shape == b.flat_grad.shape
            prev_end = 0
            for i, p in enumerate(b.params):
                off = b.offsets[i]
                assert off % 4 == 0 or off == 0
                assert off >= prev_end          # non-overlapping segments
                prev_end = off + p.numel()
                assert prev_end <= b.numel
                assert id(p) not in seen        # each param exactly once
                seen.add(id(p))
                order.append(p)
                # param data was rebound into the flat buffer, values kept
                assert p.data_ptr() >= b.flat_param.data_ptr()
                if views:
                    assert p.grad is not None
                    assert p.grad.shape == p.shape
        assert len(seen) == len(params)
        # reverse registration order across the flattened bucket sequence
        assert [id(p) for p in order] == [id(p) for p in reversed(params)]
        for p, o in zip(params, originals):
            assert torch.equal(p.detach(), o)

    check()


def _agree_worker(rank, world, port, out_dir):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    torch.distributed.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from mi355x_ddp.parallel import comm as comm_mod

        class FakeRccl:  # "succeeds" on rank 1 only
            def __init__(self, device):
                if rank == 0:
                    raise RuntimeError("injected rank-0 RCCL failure")

        orig = comm_mod.RcclCommAdapter
        comm_mod.RcclCommAdapter = FakeRccl
        try:
            c, kind = comm_mod.build_gpu_comm(torch.device("cpu"),
                                              want_mesh=False,
                                              log=lambda m: None)
        finally:
            comm_mod.RcclCommAdapter = orig
        # EVERY rank must have downgraded together — a split would hang
        assert kind == "gloo-fallback", kind
        assert isinstance(c, comm_mod.GlooComm)

        # want_mesh on a CPU host: P2pMesh setup fails on every rank and
        # the world agrees to stay on the base transport
        comm_mod.RcclCommAdapter = FakeRccl
        try:
            c2, kind2 = comm_mod.build_gpu_comm(torch.device("cpu"),
                                                want_mesh=True,
                                                log=lambda m: None)
        finally:
            comm_mod.RcclCommAdapter = orig
        assert kind2 == "gloo-fallback" and isinstance(c2, comm_mod.GlooComm)
        torch.save({"kind": kind},
                   os.path.join(out_dir, f"agree{rank}.pt"))
        torch.distributed.barrier()
    finally:
        torch.distributed.destroy_process_group()


def test_transport_agreement_on_partial_failure(tmp_path):
    import torch.multiprocessing as mp

    def _free_port():
        import socket
        with socket.socket() as s:
            s.bind(("127.0.0.1", 0))
            return s.getsockname()[1]

    mp.spawn(_agree_worker, args=(2, _free_port(), str(tmp_path)), nprocs=2,
             join=True)
    for r in range(2):
        d = torch.load(tmp_path / f"agree{r}.pt", weights_only=True)
        assert d["kind"] == "gloo-fallback"


def test_fused_sgd_rejects_multiple_groups_on_attach():
    # step() applies each flat bucket once PER GROUP; silently double-
    # stepping with two lrs would corrupt training — refuse at attach time
    import pytest
    import torch

    from mi355x_ddp.models import toy_model
    from mi355x_ddp.parallel import FusedSGD
    from mi355x_ddp.parallel.reducer import Reducer

    m1, m2 = toy_model(8, 1), toy_model(8, 1)
    opt = FusedSGD([{"params": m1.parameters(), "lr": 1e-3},
                    {"params": m2.parameters(), "lr": 1e-4}], lr=1e-3)
    red = Reducer(list(m1.parameters()) + list(m2.parameters()), comm=None)
    with pytest.raises(ValueError, match="single param_group"):
        opt.attach_reducer(red)


def test_cpp_core_not_engaged_on_cpu():
    # the C++ ReducerCore is the GPU hook path; CPU (gloo test tier) must
    # keep the Python hooks regardless of the env default
    from mi355x_ddp.parallel.reducer import Reducer

    m = nn.Linear(8, 4)
    red = Reducer(list(m.parameters()), comm=None, grad_views=True)
    assert red._core is None and red._hooks
    # skip_comm stays a plain flag on the Python path
    red.skip_comm = True
    assert red.skip_comm is True
    red.skip_comm = False
