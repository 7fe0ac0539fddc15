"""Fast train-step engines for the flagship (toy Linear DDP) workload.

The reference hot loop (single_gpu.py:21-26 / SURVEY §3.5) costs, per step:
zero_grad + fwd GEMM + loss fwd + loss bwd + two bwd GEMMs + bucket
all-reduce + SGD — with 84 B of gradients the whole thing is launch-latency
bound (SURVEY §7 hard-part 2). Three engines, in increasing aggression:

- `ToyFusedStep`: ONE kernel (fused fwd + loss grad + bwd [+ in-kernel SGD
  at world 1]); at world > 1, + one tiny all-reduce + one fused-SGD kernel.
- `GraphedToyStep`: the fused step captured in a hipGraph and replayed.
- `PersistentToyStep` (the bench default): DEFERRED runs of consecutive
  steps execute as one multi-step kernel with the weights LDS-resident;
  at world > 1 with a `P2pMeshComm` the kernel performs an in-kernel xGMI
  mesh all-reduce per step. See docs/KERNELS.md.

All engines train the SAME model parameters the generic autograd path
trains: the Reducer's flat bucket is shared state, so checkpoints and the
Trainer API are unaffected. Equivalence with the autograd path (and the
bitwise multi==single contract) is covered by tests/test_engine_gpu.py.
"""

from __future__ import annotations

from typing import Optional

import torch

from . import ops
from .models.toy import HipLinear
from .parallel.reducer import Reducer


# Process-wide capture poison: after ONE invalidated capture, any further
# capture_begin in this process is lethal on this torch/ROCm build —
# capture_begin raises "Cannot register the state during capturing stage"
# mid-registration and the partially-registered CUDAGraph's DESTRUCTOR
# then throws inside ~CUDAGraph ("The graph should be registered to the
# state") which calls std::terminate (measured: tools/recapture_probe.py,
# every cleanup variant dumps core). So: first failure -> never capture
# again in this process; all graphed engines run eager.
_CAPTURE_POISONED = False
# Graphs whose capture failed are LEAKED on purpose: destroying them can
# terminate the process (above). A handful of small host objects.
_DEAD_GRAPHS = []


def _poison_captures(graph):
    global _CAPTURE_POISONED
    _CAPTURE_POISONED = True
    _DEAD_GRAPHS.append(graph)


def _recover_failed_capture(ctx):
    """Best-effort cleanup after an INVALIDATED stream capture (e.g. a
    library allocating outside torch's allocator mid-capture — observed
    with MIOpen NHWC workspace allocs). torch.cuda.graph.__exit__ raises
    from capture_end BEFORE exiting its stream context or ending the
    allocator's route-to-pool state, leaving the thread on the capture
    stream, allocations routed to the dead graph pool, and
    hipErrorStreamCaptureInvalidated latched in HIP's thread state.
    Undo all three so the eager fallback can proceed."""
    try:
        ctx.stream_ctx.__exit__(None, None, None)
    except Exception:
        pass
    try:
        dev = torch.cuda.current_device()
        torch._C._cuda_endAllocateToPool(dev, ctx.pool[0])
        torch._C._cuda_releasePool(dev, ctx.pool[0])
    except Exception:
        pass
    try:
        # capture_begin registers the default CUDA generator and flips its
        # state to capture mode (capture_prologue); only a SUCCESSFUL
        # capture_end flips it back. Nothing on the failure path clears
        # it, so the next RNG op in the process would raise "Offset
        # increment outside graph capture". Swap in a fresh state object
        # carrying the same seed/offset (graphsafe_set_state shares the
        # new state, whose capture flag starts clear).
        dev = torch.cuda.current_device()
        dg = torch.cuda.default_generators[dev]
        fresh = torch.Generator(device=f"cuda:{dev}")
        fresh.set_state(dg.get_state())
        dg.graphsafe_set_state(fresh)
    except Exception:
        pass
    if ops.has_ext():
        ops.ext().clear_hip_errors()  # pop the latched HIP error
    torch.cuda.synchronize()


class ToyFusedStep:
    """One-kernel fwd+bwd for HipLinear(K,1) + MSE/CE, bucket all-reduce,
    fused SGD. Works for any world size (comm=None -> single process)."""

    def __init__(self, model: HipLinear, comm=None, lr: float = 1e-3,
                 use_mse: bool = True, reducer: Optional[Reducer] = None,
                 track_loss: bool = False):
        assert (isinstance(model, HipLinear) and model.out_features == 1
                and model.bias is not None), \
            "toy engines need HipLinear(K, 1) with a bias"
        self.model = model
        self.comm = comm
        self.lr = lr
        self.use_mse = use_mse
        self.reducer = reducer or Reducer(list(model.parameters()), comm=comm)
        assert len(self.reducer.buckets) == 1, "toy model is one bucket"
        b = self.reducer.buckets[0]
        self.flat_param = b.flat_param
        self.flat_grad = b.flat_grad
        _, widx = self.reducer._param_index[model.weight]
        _, bidx = self.reducer._param_index[model.bias]
        self.w_off = b.offsets[widx]
        self.b_off = b.offsets[bidx]
        self.loss_out = (torch.zeros((), device=self.flat_param.device)
                         if track_loss else torch.Tensor())

    def step(self, x: torch.Tensor, t: torch.Tensor) -> None:
        if self.comm is None:
            # single-process: one launch = one full training step
            # (fwd + loss grad + bwd + SGD fused; no collective to wait for)
            ops.ext().toy_fused_fwd_bwd(x, t, self.flat_param, self.flat_grad,
                                        self.loss_out, self.use_mse,
                                        self.w_off, self.b_off, self.lr)
            return
        ops.ext().toy_fused_fwd_bwd(x, t, self.flat_param, self.flat_grad,
                                    self.loss_out, self.use_mse,
                                    self.w_off, self.b_off, 0.0)
        self.comm.all_reduce_avg_inline(self.flat_grad)
        ops.ext().sgd_flat(self.flat_param, self.flat_grad, self.lr, True)


class PersistentToyStep(ToyFusedStep):
    """Deferred-launch engine: runs of consecutive steps execute as ONE
    multi-step kernel launch (`toy_multistep`), weights resident in LDS
    across steps. Works at world 1 (no communicator) and, with a
    `P2pMeshComm`, at any world size — the kernel then performs one
    in-kernel xGMI mesh all-reduce per step (`toy_multistep_mesh`).

    `step(x, t)` defers when the incoming batch is the next contiguous
    [B, K] slice of the same device buffer (the device-resident epoch
    shard, bench.py DeviceData); any other batch flushes the pending run
    and starts a new one. Per-step arithmetic is bitwise-identical to the
    single-step fused kernel, so deferral changes WHEN work is launched,
    never what is computed. Callers must call `flush()` before
    synchronizing the stream for timing/reading params — bench.py does
    before every barrier. Deferred-run length is capped by `max_defer`.

    With track_loss=True, loss_out holds the LAST executed step's loss.
    """

    def __init__(self, *args, max_defer: int = 1024, **kwargs):
        super().__init__(*args, **kwargs)
        # world-1 (no comm), or a mesh-capable comm: the in-kernel xGMI
        # mesh exchange keeps SGD in-kernel at any world size
        self._mesh = getattr(self.comm, "_mesh", None)
        assert self.comm is None or self._mesh is not None, \
            "PersistentToyStep needs world 1 or a P2pMeshComm"
        self.max_defer = max_defer
        self._install_fast_path()

    def _install_fast_path(self) -> None:
        """step/flush as closures over cell variables: the per-step cost is
        two data_ptr() calls and a shape compare, no attribute traffic —
        this loop runs a few hundred ns behind a kernel that takes ~1.3 us
        per step, so host bookkeeping is the wall-clock bound."""
        ext_mod = ops.ext() if ops.has_ext() else None
        flat_param = self.flat_param
        loss_out = self.loss_out
        use_mse, w_off, b_off = self.use_mse, self.w_off, self.b_off
        lr, max_defer = self.lr, self.max_defer
        mesh = self._mesh

        self.launch_count = 0  # kernel launches issued (bench transparency)

        def launch(xall, tall, B):
            """One deferred-run launch: world-1 multistep, or the mesh
            variant with an in-kernel all-reduce per step."""
            self.launch_count += 1
            if mesh is None:
                ext_mod.toy_multistep(xall, tall, flat_param, loss_out,
                                      use_mse, w_off, b_off, lr, B)
            else:
                ext_mod.toy_multistep_mesh(xall, tall, flat_param, loss_out,
                                           use_mse, w_off, b_off, lr, B,
                                           mesh)

        if mesh is None:
            eager = ToyFusedStep.step.__get__(self)
        else:
            def eager(x, t):  # single-step via the same mesh kernel (S=1):
                # keeps the per-step collective order identical on all ranks
                launch(x.contiguous(), t.contiguous(), x.shape[0])
        x0 = t0 = None
        shape = None
        count = nx = nt = bk = bt = 0

        def step(x: torch.Tensor, t: torch.Tensor) -> None:
            nonlocal x0, t0, count, nx, nt, bk, bt, shape
            if s_next != s_start:  # a shard-bound run is pending: order it
                flush()
            if count:
                # fast path: the next contiguous slice of the pending run
                if (x.data_ptr() == nx and t.data_ptr() == nt
                        and x.shape == shape):
                    count += 1
                    nx += bk
                    nt += bt
                    if count >= max_defer:
                        flush()
                    return
                flush()
            if not (x.is_cuda and x.is_contiguous() and t.is_contiguous()):
                eager(x, t)
                return
            x0, t0, count = x, t, 1
            shape = x.shape
            bk = x.numel() * x.element_size()
            bt = t.numel() * t.element_size()
            nx = x.data_ptr() + bk
            nt = t.data_ptr() + bt

        # shard-bound path: bind the epoch shard once, then step by batch
        # INDEX — no per-step tensor-view construction at all. Used by
        # bench.py's device-resident data path; step(x, t) remains the
        # generic API.
        shard_x = shard_t = None
        sbatch = 0
        s_start = s_next = 0  # pending run = batch indices [s_start, s_next)

        def bind_shard(xs: torch.Tensor, ts: torch.Tensor, batch: int) -> None:
            nonlocal shard_x, shard_t, sbatch, s_start, s_next
            flush()
            assert xs.is_cuda and xs.is_contiguous() and ts.is_contiguous()
            shard_x, shard_t, sbatch = xs, ts, batch
            s_start = s_next = 0

        def step_shard(i: int) -> None:
            nonlocal s_start, s_next
            if i == s_next:
                s_next += 1
                if s_next - s_start >= max_defer:
                    flush()
                return
            flush()
            s_start, s_next = i, i + 1

        def flush() -> None:
            nonlocal x0, t0, count, s_start, s_next
            if count:
                xx, tt, n = x0, t0, count
                x0 = t0 = None
                count = 0
                if n == 1:
                    launch(xx, tt, xx.shape[0])  # S=1: bitwise == fused
                else:
                    B, K = xx.shape
                    xall = xx.as_strided((n * B, K), (K, 1))
                    tall = tt.as_strided((n * B,) + tt.shape[1:],
                                         (tt.stride(0),) + tt.stride()[1:])
                    launch(xall, tall, B)
            if s_next > s_start:
                lo, n = s_start * sbatch, (s_next - s_start) * sbatch
                s_start = s_next
                launch(shard_x[lo:lo + n], shard_t[lo:lo + n], sbatch)

        self.step = step
        self.flush = flush
        self.bind_shard = bind_shard
        self.step_shard = step_shard


class GraphedAutogradStep:
    """The GENERIC training step — forward, loss, autograd backward with
    the reducer hooks, bucket all-reduce, fused SGD — captured in ONE
    hipGraph and replayed (torch "whole-network capture").

    This is what makes the arbitrary-model path fast on MI355X: eager, the
    step pays ~10 kernel launches plus autograd-engine and Python dispatch
    overhead (~100 us for the toy model, profiles/README.md r02); replayed,
    it pays two tiny input copies plus the graph-replay floor (~10-16 us,
    MI355X_MICROARCH price list). The step is capture-safe BY CONSTRUCTION
    here: gradients are views into the reducer's static flat buckets,
    FusedSGD updates the static flat pairs, and the reducer's bucket
    collectives are launched with stream-event edges capture records.
    The wrapped model's reducer must use PYTHON hooks
    (DDP(..., cpp_hooks=False)): a C++ node post-hook — even a no-op one —
    segfaults hipStreamEndCapture on this torch/ROCm build (bisected on
    hardware, tools/capture_bisect2.py). Hooks never execute during
    replay, so the captured path loses nothing vs the C++ core.

    Deferred multi-step replay: besides the single-step `step(x, t)` API,
    the engine speaks the shard-bound protocol (`bind_shard`/`step_shard`/
    `flush`, same contract as PersistentToyStep): runs of consecutive
    batch indices execute as replays of MULTI-STEP graphs — a G-step graph
    amortizes the replay floor and the input copies over G steps, leaving
    the in-graph kernel-boundary cost (~1.2-1.5 us per dependent kernel)
    as the bound. Runs decompose greedily over `chunk_sizes` graphs, each
    captured once on first use; warmup covers the steady-state sizes so
    captures stay out of timed regions.

    Requirements: static shapes (same batch size every step) and a model
    without data-dependent control flow — the reference workloads (toy
    Linear, ResNet-50) qualify. Autocast regions inside the step must use
    cache_enabled=False (torch's AMP+graphs rule — the cast cache frees
    tensors between iterations and invalidates the capture). Capture
    failure falls back to eager with a warning; results are unchanged
    either way.
    """

    def __init__(self, model, loss_fn, optimizer, finalize=None,
                 warmup_steps: int = 3, zero_grad: bool = False,
                 chunk_sizes=(64, 8, 1)):
        self.model = model            # DDP-wrapped or bare
        self.loss_fn = loss_fn
        self.optimizer = optimizer
        self._finalize = finalize or (lambda: None)
        self.warmup_steps = warmup_steps
        # zero_grad=True: the optimizer does not fold grad zeroing into its
        # step (i.e. it is not a bucket-attached FusedSGD) — zero INSIDE
        # the captured step, set_to_none=False so the grad buffers stay
        # static (a capture requirement, and it keeps the bucket views).
        self._zero = zero_grad
        self.chunk_sizes = tuple(sorted(set(chunk_sizes) | {1},
                                        reverse=True))
        self.max_defer = self.chunk_sizes[0]
        self._graphs = {}             # G -> (graph, xbuf, tbuf)
        self._broken = False          # capture failed: eager forever
        self._warmed = False
        self._bshape = None           # (x batch shape, t batch shape)
        self.loss = None              # last captured loss tensor
        # shard-bound state
        self._shard_x = self._shard_t = None
        self._sbatch = 0
        self._s_start = self._s_next = 0

    def _eager_step(self, x, t):
        if self._zero:
            self.optimizer.zero_grad(set_to_none=False)
        loss = self.loss_fn(self.model(x), t)
        loss.backward()
        self._finalize()
        self.optimizer.step()
        return loss

    def _warm(self, x, t):
        """torch's capture recipe: a few eager steps on a side stream
        before the first capture (allocator + autograd engine initialize
        lazily-created state off-capture). These steps train for real —
        on the first batch, like any warmup."""
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for _ in range(self.warmup_steps):
                self._eager_step(x, t)
        torch.cuda.current_stream().wait_stream(s)
        torch.cuda.synchronize()
        self._warmed = True

    def _get_graph(self, G):
        """The G-step graph (captured on first use). Returns None when
        capture is broken; xbuf/tbuf hold G consecutive batches."""
        got = self._graphs.get(G)
        if got is not None or self._broken:
            return got
        if _CAPTURE_POISONED:
            self._broken = True  # an earlier capture in this process
            return None          # failed: capturing again is lethal
        B = self._bshape[0][0]
        xbuf = torch.empty((G * B,) + self._bshape[0][1:],
                           dtype=self._bdtype[0], device=self._bdev)
        tbuf = torch.empty((G * B,) + self._bshape[1][1:],
                           dtype=self._bdtype[1], device=self._bdev)
        g = torch.cuda.CUDAGraph()
        ctx = torch.cuda.graph(g)
        try:
            with ctx:
                for i in range(G):
                    self.loss = self._eager_step(xbuf[i * B:(i + 1) * B],
                                                 tbuf[i * B:(i + 1) * B])
            self._graphs[G] = (g, xbuf, tbuf)
            return self._graphs[G]
        except Exception as e:
            import warnings
            warnings.warn(
                f"[mi355x_ddp] whole-step hipGraph capture failed ({e!r}); "
                "GraphedAutogradStep running eager (correct, slower; no "
                "further captures will be attempted in this process)",
                RuntimeWarning, stacklevel=2)
            self._broken = True
            _poison_captures(g)
            _recover_failed_capture(ctx)
            return None

    def _note_shapes(self, x, t):
        if self._bshape is None:
            self._bshape = (x.shape, t.shape)
            self._bdtype = (x.dtype, t.dtype)
            self._bdev = x.device

    def step(self, x, t):
        """Single-step API (Trainer): one G=1 graph replay per call."""
        self.flush()
        self._note_shapes(x, t)
        if not self._broken and x.shape != self._bshape[0]:
            self._eager_step(x, t)  # ragged final batch: run it eager
            return
        if not self._warmed:
            self._warm(x, t)
        got = self._get_graph(1)
        if got is None:
            self._eager_step(x, t)
            return
        g, xbuf, tbuf = got
        xbuf.copy_(x.reshape(xbuf.shape), non_blocking=True)
        tbuf.copy_(t.reshape(tbuf.shape), non_blocking=True)
        g.replay()

    # -- shard-bound protocol (same contract as PersistentToyStep) -------
    def bind_shard(self, xs, ts, batch):
        self.flush()
        assert xs.is_cuda and xs.is_contiguous() and ts.is_contiguous()
        self._shard_x, self._shard_t, self._sbatch = xs, ts, batch
        self._s_start = self._s_next = 0
        if (self._bshape is not None
                and (xs[:batch].shape, ts[:batch].shape) != self._bshape):
            # batch shape changed across binds: the captured graphs read
            # fixed-shape static buffers — drop them and recapture lazily
            # at the new shape (correct, pays one recapture)
            torch.cuda.synchronize()
            self._graphs.clear()
            self._bshape = None
        self._note_shapes(xs[:batch], ts[:batch])

    def step_shard(self, i):
        if i == self._s_next:
            self._s_next += 1
            if self._s_next - self._s_start >= self.max_defer:
                self.flush()
            return
        self.flush()
        self._s_start, self._s_next = i, i + 1

    def chunks(self, n):
        """Greedy decomposition of an n-step run over chunk_sizes graphs
        (largest-first; chunk_sizes always contains 1, so any n is
        covered exactly). Pure logic — property-tested on CPU."""
        out = []
        while n:
            for G in self.chunk_sizes:
                if G <= n:
                    break
            out.append(G)
            n -= G
        return out

    def flush(self):
        if self._s_next <= self._s_start:
            return
        lo, n = self._s_start, self._s_next - self._s_start
        self._s_start = self._s_next
        B = self._sbatch
        if not self._warmed:
            self._warm(self._shard_x[lo * B:(lo + 1) * B],
                       self._shard_t[lo * B:(lo + 1) * B])
        for G in self.chunks(n):
            s, e = lo * B, (lo + G) * B
            got = self._get_graph(G)
            if got is None:  # capture broken: eager the rest of the run
                for i in range(lo, lo + n):
                    self._eager_step(self._shard_x[i * B:(i + 1) * B],
                                     self._shard_t[i * B:(i + 1) * B])
                return
            g, xbuf, tbuf = got
            xbuf.copy_(self._shard_x[s:e], non_blocking=True)
            tbuf.copy_(self._shard_t[s:e], non_blocking=True)
            g.replay()
            lo += G
            n -= G


class GraphedToyStep(ToyFusedStep):
    """ToyFusedStep captured in a hipGraph.

    The input batch lives in static device buffers; `step` copies the batch
    view into them (D2D, tiny) and replays the graph. RCCL collectives are
    capturable on ROCm (stream capture follows the comm-stream event edges).
    """

    def __init__(self, *args, **kwargs):
        super().__init__(*args, **kwargs)
        self._graph = None
        self._x_static = None
        self._t_static = None

    def _capture(self, x: torch.Tensor, t: torch.Tensor) -> None:
        self._x_static = torch.empty_like(x)
        self._t_static = torch.empty_like(t)
        self._x_static.copy_(x)
        self._t_static.copy_(t)
        # warm up collectives/kernels outside capture first
        super().step(self._x_static, self._t_static)
        torch.cuda.synchronize()
        if _CAPTURE_POISONED:
            self._graph = False  # earlier failed capture: stay eager
            return
        g = torch.cuda.CUDAGraph()
        ctx = torch.cuda.graph(g)
        try:
            with ctx:
                super().step(self._x_static, self._t_static)
            self._graph = g
        except Exception as e:  # capture unsupported -> eager fallback
            import warnings
            warnings.warn(
                f"[mi355x_ddp] hipGraph capture failed ({e!r}); "
                "GraphedToyStep running eager (correct, slower)",
                RuntimeWarning, stacklevel=2)
            self._graph = False
            _poison_captures(g)
            _recover_failed_capture(ctx)

    def step(self, x: torch.Tensor, t: torch.Tensor) -> None:
        if self._graph is None:
            self._capture(x, t)
            return
        if self._graph is False:
            super().step(x, t)
            return
        self._x_static.copy_(x, non_blocking=True)
        self._t_static.copy_(t, non_blocking=True)
        self._graph.replay()
