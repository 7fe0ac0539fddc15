"""Flagship benchmark: toy Linear DDP training step on 1..8 MI355X GPUs.

Measures the BASELINE.json metric — samples/sec (whole node) and step-time
for the reference workload (Linear(20,1), batch 32/rank, dataset 2048,
SURVEY §6) — through the native engine: hand-written MFMA kernels, flat
gradient bucket, RCCL all-reduce over xGMI, fused SGD.

Single process:          python bench.py [--gpus 1] [--steps K] [--warmup W]
Multi-GPU (driver runs): python -m torch.distributed.run --nnodes=1 \
    --nproc-per-node N --master-addr 127.0.0.1 --master-port P \
    bench.py --gpus N --steps K --warmup W

Engines (--engine): persistent (default; S steps per kernel launch at
world 1, falls back to fused when a communicator exists), fused (one
fused kernel + collective + fused SGD per step), graph (fused step
captured in a hipGraph), autograd (the generic Trainer/DDP path with
per-param hooks — the path arbitrary models take).
Synthetic data (random, reference shapes), random-init weights; fp32 (the
reference's precision) or --dtype bf16 (BASELINE config 2).
"""

from __future__ import annotations

import argparse
import json
import os
import sys
import time

import torch


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=2000)
    p.add_argument("--warmup", type=int, default=200)
    p.add_argument("--engine",
                   choices=["persistent", "fused", "graph", "autograd",
                            "autograd-graph"],
                   default="persistent",
                   help="persistent = multi-step kernel at world 1 "
                        "(falls back to fused when a comm exists); "
                        "autograd = the generic hook/reducer path; "
                        "autograd-graph = the generic path captured in "
                        "one hipGraph and replayed (whole-step capture)")
    p.add_argument("--batch", type=int, default=32, help="batch per rank")
    p.add_argument("--lr", type=float, default=1e-3)
    p.add_argument("--dataset", type=int, default=2048)
    p.add_argument("--p50-probes", type=int, default=64)
    p.add_argument("--dtype", choices=["fp32", "bf16"], default="fp32",
                   help="compute dtype (fp32 = the reference's; bf16 = "
                        "BASELINE.json config 2)")
    p.add_argument("--loss", choices=["mse", "ce"], default="mse",
                   help="mse (default: the reference's multinode loss; "
                        "actually trains) or ce (the reference's stage-1 "
                        "loss — degenerate over one logit: zero gradients, "
                        "SURVEY §2.1)")
    return p.parse_args()


class DeviceData:
    """Device-resident dataset + per-epoch shard gather (the framework's
    fast data path: one 164 KB upload, per-epoch permute+gather on device,
    per-step slicing is free)."""

    def __init__(self, n, rank, world, batch, device, seed=1234,
                 dtype=torch.float32):
        g = torch.Generator().manual_seed(seed)
        self.X = torch.rand(n, 20, generator=g).to(device=device, dtype=dtype)
        self.T = torch.rand(n, 1, generator=g).to(device=device, dtype=dtype)
        self.rank, self.world, self.batch = rank, world, batch
        self.per_rank = n // world
        self.steps_per_epoch = self.per_rank // batch
        self.device = device
        self._cuda = device.type == "cuda"
        self._epoch = -1
        self._gen = (torch.Generator(device=device) if self._cuda
                     else torch.Generator())
        # MI355X_EPOCH_SIDE_STREAM=1 gathers epoch shards on a side stream
        # one epoch ahead; measured SLOWER than the inline default (A/B on
        # one box: 9.9 vs 10.8 M samples/s) — the cross-stream allocation
        # + event bookkeeping costs more than the ~25 us/epoch it hides.
        use_side = os.environ.get("MI355X_EPOCH_SIDE_STREAM", "0") == "1"
        self._side = torch.cuda.Stream() if (self._cuda and use_side) else None
        self._pending = {}  # epoch -> (xs, ts, ready_event)
        self.bound_epoch = -1  # used by the shard-bound engine path
        self.bound_block = -1  # used by the epoch-block engine path
        self._fused_shard = os.environ.get("MI355X_SHUFFLE", "") != "randperm"

    def _gather(self, epoch):
        if self._cuda and self._fused_shard:
            # ONE kernel: epoch-seeded bijective permutation computed
            # inline per row + linear copy (replaces randperm's radix sort
            # + two index_selects; MI355X_SHUFFLE=randperm for the old path)
            from mi355x_ddp import ops
            xs, ts = ops.ext().epoch_shard(self.X, self.T, 1000 + epoch,
                                           self.rank, self.world)
            return xs, ts
        # epoch-seeded permutation generated ON DEVICE (identical on every
        # rank for a given epoch; no host round trip)
        self._gen.manual_seed(1000 + epoch)
        perm = torch.randperm(self.X.shape[0], generator=self._gen,
                              device=self.device)
        shard = perm[self.rank::self.world][: self.per_rank]
        return self.X[shard].contiguous(), self.T[shard].contiguous()

    def _prep(self, epoch):
        if self._side is None:
            self._pending[epoch] = (*self._gather(epoch), None)
            return
        with torch.cuda.stream(self._side):
            xs, ts = self._gather(epoch)
            ev = torch.cuda.Event()
            ev.record(self._side)
        self._pending[epoch] = (xs, ts, ev)

    def shard_for(self, epoch):
        """The gathered epoch shard (xs, ts), for shard-bound engines —
        no batch views are built on this path."""
        if epoch not in self._pending:
            self._prep(epoch)
        xs, ts, ev = self._pending.pop(epoch)
        if ev is not None:
            main = torch.cuda.current_stream()
            main.wait_event(ev)
            xs.record_stream(main)
            ts.record_stream(main)
        self._prep(epoch + 1)
        return xs, ts

    def shards_for_block(self, e0, epochs):
        """One contiguous buffer pair covering epochs [e0, e0+epochs):
        block e is exactly shard_for(e0+e)'s content (bitwise). Fused
        path: ONE gather kernel for the whole block (epoch_shard_multi) —
        this is what lets the persistent engine's deferral span epoch
        boundaries (one multistep launch per block instead of per epoch)."""
        if self._cuda and self._fused_shard:
            from mi355x_ddp import ops
            return ops.ext().epoch_shard_multi(self.X, self.T, 1000 + e0,
                                               epochs, self.rank, self.world)
        xs, ts = zip(*(self._gather(e0 + e) for e in range(epochs)))
        return torch.cat(xs), torch.cat(ts)

    def batch_for(self, step):
        epoch, s = divmod(step, self.steps_per_epoch)
        if epoch != self._epoch:
            if epoch not in self._pending:
                self._prep(epoch)
            xs, ts, ev = self._pending.pop(epoch)
            if ev is not None:
                main = torch.cuda.current_stream()
                main.wait_event(ev)
                # shards are allocated on the side stream but consumed on the
                # main stream: mark the usage so the caching allocator only
                # reuses their blocks after main-stream kernels finish
                xs.record_stream(main)
                ts.record_stream(main)
            self._prep(epoch + 1)  # prefetch next epoch on the side stream
            # one C-side split call instead of 2*steps Python slicings
            self._views = list(zip(torch.split(xs, self.batch),
                                   torch.split(ts, self.batch)))
            self._epoch = epoch
        return self._views[s]


def drive_shard_bound(data, spe, batch, eb, bind, step_shard, start, n):
    """Drive a shard-bound engine for global steps [start, start+n):
    (re)bind the covering shard — one epoch, or an eb-epoch block — when
    stepping crosses into a new one, then step by batch INDEX within the
    bound buffer. Pure index logic, unit-tested on CPU with a recording
    fake engine (tests/test_bench_contract.py) because the driver's
    8-GPU scale run exercises shapes (spe=8, eb=128) that never run in
    the 1-GPU harness. Mid-run starts (the timed region begins at
    step=warmup, rarely a block boundary) re-enter the current partial
    block/epoch via the engine's non-sequential step_shard restart."""
    if eb == 1:
        cur = data.bound_epoch
        for s in range(start, start + n):
            e, i = divmod(s, spe)
            if e != cur:
                xs, ts = data.shard_for(e)
                bind(xs, ts, batch)
                data.bound_epoch = cur = e
            step_shard(i)
        return
    cur = data.bound_block
    for s in range(start, start + n):
        e, i = divmod(s, spe)
        b = e // eb
        if b != cur:
            xs, ts = data.shards_for_block(b * eb, eb)
            bind(xs, ts, batch)
            data.bound_block = cur = b
        step_shard((e - b * eb) * spe + i)


def build_engine(kind, comm, lr, device, dtype=torch.float32, use_mse=True):
    from mi355x_ddp.engine import (GraphedToyStep, PersistentToyStep,
                                   ToyFusedStep)
    from mi355x_ddp.models import toy_model
    from mi355x_ddp.parallel import DDP, FusedSGD
    from mi355x_ddp import ops

    torch.manual_seed(4242)  # same init on every rank
    model = toy_model(20, 1).to(device=device, dtype=dtype)

    noflush = lambda: None  # noqa: E731
    if kind in ("autograd", "autograd-graph"):
        # capture requires Python hooks (a C++ node post-hook segfaults
        # hipStreamEndCapture — reducer.py); replay runs no hooks at all
        engine = DDP(model, comm=comm,
                     cpp_hooks=False if kind == "autograd-graph" else None)
        opt = FusedSGD(model.parameters(), lr=lr)
        opt.attach_reducer(engine.reducer)

        loss_fn = ops.mse_loss if use_mse else ops.cross_entropy

        if kind == "autograd-graph" and device.type == "cuda":
            from mi355x_ddp.engine import GraphedAutogradStep
            gs = GraphedAutogradStep(engine, loss_fn, opt,
                                     finalize=engine.finalize_backward)
            return gs.step, gs.flush, gs  # shard-bound protocol supported

        def step(x, t):
            loss = loss_fn(engine(x), t)
            loss.backward()
            engine.finalize_backward()
            opt.step()
        return step, noflush, None

    if (kind == "persistent" and comm is not None
            and getattr(comm, "_mesh", None) is None):
        kind = "fused"  # without a mesh comm, multi-step is world-1 only
    cls = {"persistent": PersistentToyStep, "graph": GraphedToyStep,
           "fused": ToyFusedStep}[kind]
    eng = cls(model, comm=comm, lr=lr, use_mse=use_mse)
    if comm is not None:
        eng.reducer.broadcast_params(root=0)
    return eng.step, getattr(eng, "flush", noflush), eng


def main():
    args = parse_args()
    rank = int(os.environ.get("RANK", 0))
    world = int(os.environ.get("WORLD_SIZE", 1))
    local_rank = int(os.environ.get("LOCAL_RANK", 0))

    use_cuda = torch.cuda.is_available()
    if os.environ.get("MI355X_FORCE_DEV0") == "1":
        local_rank = 0  # test-only: run a multi-rank world on ONE device
    device = torch.device("cuda", local_rank) if use_cuda else torch.device("cpu")
    if use_cuda:
        torch.cuda.set_device(device)

    comm = None
    comm_kind = "none"
    dist = torch.distributed
    if world > 1:
        # gloo for rendezvous/barriers only; gradient bytes ride RcclComm
        dist.init_process_group("gloo", rank=rank, world_size=world)
        from mi355x_ddp.parallel.comm import GlooComm, build_gpu_comm
        if use_cuda:
            # hang-safe transport ladder: RCCL (-> gloo), + the xGMI mesh
            # when every rank agrees it set up AND cross-validated; any
            # downgrade happens on ALL ranks together so collectives never
            # mismatch (mi355x_ddp.parallel.comm.build_gpu_comm)
            comm, comm_kind = build_gpu_comm(
                device, want_mesh=os.environ.get("MI355X_P2P", "1") != "0",
                log=lambda m: print(m, file=sys.stderr, flush=True))
        else:
            comm = GlooComm()
            comm_kind = "gloo-cpu"

    def barrier():
        if world > 1:
            dist.barrier()
        if use_cuda:
            torch.cuda.synchronize()

    dtype = torch.bfloat16 if args.dtype == "bf16" else torch.float32
    data = DeviceData(args.dataset, rank, world, args.batch, device,
                      dtype=dtype if use_cuda else torch.float32)
    spe = data.steps_per_epoch
    if spe < 1:
        raise SystemExit(
            f"--dataset {args.dataset} yields {data.per_rank} samples/rank "
            f"at world {world}: smaller than one --batch {args.batch}")

    step_fn = flush_fn = engine_obj = None
    eng_bind = eng_step_shard = None

    def make_engine():
        nonlocal step_fn, flush_fn, engine_obj, eng_bind, eng_step_shard
        step_fn, flush_fn, engine_obj = build_engine(
            args.engine if use_cuda else "autograd", comm, args.lr, device,
            dtype if use_cuda else torch.float32, use_mse=args.loss == "mse")
        # shard-bound fast path: bind each epoch's shard once, then drive
        # the engine by batch INDEX (no per-step view construction)
        eng_bind = getattr(engine_obj, "bind_shard", None)
        eng_step_shard = getattr(engine_obj, "step_shard", None)
        data.bound_epoch = -1
        data.bound_block = -1

    make_engine()

    # Epoch-block binding: gather MI355X_EPOCH_BLOCK epochs' shards in ONE
    # kernel (epoch_shard_multi) and bind them as one shard, so the
    # engine's deferral — and therefore the multistep kernel launch —
    # spans epoch boundaries. Default fills the engine's deferral window
    # (max_defer steps). Needs whole batches per epoch (block row layout)
    # and the fused shard path; MI355X_EPOCH_BLOCK=1 restores per-epoch.
    eb = 1
    if (eng_bind is not None and use_cuda and spe > 0
            and data.per_rank == spe * args.batch and data._fused_shard):
        md = getattr(engine_obj, "max_defer", 1024)
        eb_env = os.environ.get("MI355X_EPOCH_BLOCK")
        eb = max(1, int(eb_env) if eb_env else md // spe)

    def run_steps(start, n):
        if eng_bind is None:
            for s in range(start, start + n):
                x, t = data.batch_for(s)
                step_fn(x, t)
            return
        drive_shard_bound(data, spe, args.batch, eb, eng_bind,
                          eng_step_shard, start, n)

    def comm_check():
        if comm is not None and hasattr(comm, "check"):
            comm.check()  # raises if a mesh all-reduce ever timed out

    # -- warmup (untimed) -------------------------------------------------
    barrier()  # align ranks before the first mesh-synchronized step
    run_steps(0, args.warmup)
    flush_fn()
    barrier()
    # Post-warmup transport health check. If the mesh misbehaved during
    # warmup on ANY rank (timeout / cross-validation mismatch), EVERY rank
    # downgrades to the base transport together (mismatched transports
    # would hang) and warmup is redone on the safe path.
    ok = True
    try:
        comm_check()
        if comm is not None and hasattr(comm, "validate"):
            comm.validate()
    except Exception as e:
        print(f"[bench] transport health check failed: {e!r}",
              file=sys.stderr, flush=True)
        ok = False
    if os.environ.get("MI355X_TEST_FAIL_HEALTH") == "1":
        ok = False  # test-only: exercise the downgrade + re-warmup path
    if world > 1:
        from mi355x_ddp.parallel.comm import all_ranks_agree
        if not all_ranks_agree(ok):
            if hasattr(comm, "base"):
                comm = comm.base
                comm_kind = comm_kind.split("+", 1)[-1] + "(post-warmup)"
                print(f"[bench] downgraded to {comm_kind}; redoing warmup",
                      file=sys.stderr, flush=True)
            make_engine()
            run_steps(0, args.warmup)
            flush_fn()
            barrier()
    elif not ok:
        raise RuntimeError("transport health check failed at world 1")

    # -- timed region: exactly K steps (any deferred launches are flushed
    #    INSIDE the bracket — all K steps' work executes before the
    #    closing barrier+synchronize) ------------------------------------
    launches0 = getattr(engine_obj, "launch_count", None)
    t0 = time.perf_counter()
    run_steps(args.warmup, args.steps)
    flush_fn()
    barrier()
    elapsed = time.perf_counter() - t0
    launches_timed = (getattr(engine_obj, "launch_count", 0) - launches0
                      if launches0 is not None else None)
    comm_check()

    # max over ranks (gloo all-reduce of the scalar)
    if world > 1:
        e = torch.tensor([elapsed])
        dist.all_reduce(e, op=dist.ReduceOp.MAX)
        elapsed = float(e)

    # -- p50 step time (instrumented separately, outside the headline
    #    region: per-step sync would serialize the pipeline) --------------
    import statistics
    probes = []
    for s in range(args.p50_probes):
        x, t = data.batch_for(s)
        if use_cuda:
            torch.cuda.synchronize()
        t1 = time.perf_counter()
        step_fn(x, t)
        flush_fn()  # deferred engines: probe = true single-step latency
        if use_cuda:
            torch.cuda.synchronize()
        probes.append(time.perf_counter() - t1)
    p50_ms = statistics.median(probes) * 1e3 if probes else None
    if world > 1 and p50_ms is not None:
        e = torch.tensor([p50_ms])
        dist.all_reduce(e, op=dist.ReduceOp.MAX)
        p50_ms = float(e)

    samples_per_sec = world * args.batch * args.steps / elapsed
    if rank == 0:
        result = {
            "metric": "samples/sec (whole node), toy Linear DDP",
            "value": samples_per_sec,
            "unit": "samples/s",
            "n_gpus": world if use_cuda else 0,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": elapsed / args.steps * 1e3,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,  # the reference publishes no numbers
            "dtype": args.dtype if use_cuda else "fp32",
            "data": "synthetic",
            "config": {
                "model": "Linear(20,1)",
                "global_batch": world * args.batch,
                "in_features": 20,
                "dataset_size": args.dataset,
                "parallelism": f"dp{world}",
                "comm": comm_kind,
                "engine": ("autograd-cpu" if not use_cuda else
                           args.engine if engine_obj is None else
                           {"PersistentToyStep": "persistent",
                            "GraphedToyStep": "graph",
                            "GraphedAutogradStep": "autograd-graph",
                            "ToyFusedStep": "fused"}[
                               type(engine_obj).__name__]),
                "loss": args.loss,
                "epoch_block": eb,
                "p50_step_ms": p50_ms,
                "launches_timed": launches_timed,
            },
        }
        # Transparency for short runs (the driver defaults to --steps 20):
        # the deferred engine amortizes ~2.7 us of launch overhead over up
        # to max_defer steps, so a timed region spanning only a couple of
        # launches underestimates the steady state (measured at 30k steps,
        # profiles/README.md r01q). Say so in the JSON instead of leaving
        # an apparent builder-vs-driver discrepancy.
        if args.steps < 2000:
            result["config"]["steps_note"] = (
                f"short run: {args.steps} timed steps across "
                f"{launches_timed if launches_timed is not None else '?'} "
                "kernel launch(es) — launch overhead not amortized; "
                "steady-state throughput (>=30000 steps) is higher "
                "(see profiles/README.md)")
        print(json.dumps(result))
    if world > 1:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
