"""Fast train-step engines for the flagship (toy Linear DDP) workload.

The reference hot loop (single_gpu.py:21-26 / SURVEY §3.5) costs, per step:
zero_grad + fwd GEMM + loss fwd + loss bwd + two bwd GEMMs + bucket
all-reduce + SGD — with 84 B of gradients the whole thing is launch-latency
bound (SURVEY §7 hard-part 2). `ToyFusedStep` collapses it to:

    1 kernel  (fused fwd + loss grad + bwd, writes grads into the bucket)
  [ + 1 RCCL all-reduce when world > 1 ]
    1 kernel  (fused SGD over the flat bucket, zeroes grads)

`GraphedToyStep` additionally captures the whole step (including the RCCL
collective) into a hipGraph once and replays it per step — HIP streams and
graphs instead of per-launch host dispatch.

Both engines run the SAME model parameters the generic autograd path
trains: the Reducer's flat bucket is shared state, so checkpoints and the
Trainer API are unaffected. Numerics equivalence with the autograd path is
covered by tests/test_engine_gpu.py.
"""

from __future__ import annotations

from typing import Optional

import torch

from . import ops
from .models.toy import HipLinear
from .parallel.reducer import Reducer


class ToyFusedStep:
    """One-kernel fwd+bwd for HipLinear(K,1) + MSE/CE, bucket all-reduce,
    fused SGD. Works for any world size (comm=None -> single process)."""

    def __init__(self, model: HipLinear, comm=None, lr: float = 1e-3,
                 use_mse: bool = True, reducer: Optional[Reducer] = None,
                 track_loss: bool = False):
        assert isinstance(model, HipLinear) and model.out_features == 1
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
    """World-1 engine: runs of consecutive steps execute as ONE multi-step
    kernel launch (`toy_multistep`), weights resident in LDS across steps.

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
        assert self.comm is None, \
            "PersistentToyStep is the world-1 path (in-kernel SGD)"
        self.max_defer = max_defer
        self._x0 = None        # first batch of the pending run (keeps storage)
        self._t0 = None
        self._count = 0
        self._nx = 0           # expected data_ptr of the next contiguous batch
        self._nt = 0
        self._bk = 0           # batch strides in bytes
        self._bt = 0
        self._shape = None

    def step(self, x: torch.Tensor, t: torch.Tensor) -> None:
        if self._count > 0:
            # fast path: the next contiguous slice of the pending run
            if (x.data_ptr() == self._nx and t.data_ptr() == self._nt
                    and x.shape == self._shape):
                self._count += 1
                self._nx += self._bk
                self._nt += self._bt
                if self._count >= self.max_defer:
                    self.flush()
                return
            self.flush()
        if not (x.is_cuda and x.is_contiguous() and t.is_contiguous()):
            super().step(x, t)
            return
        self._x0, self._t0, self._count = x, t, 1
        self._shape = x.shape
        self._bk = x.numel() * x.element_size()
        self._bt = t.numel() * t.element_size()
        self._nx = x.data_ptr() + self._bk
        self._nt = t.data_ptr() + self._bt

    def flush(self) -> None:
        if self._count == 0:
            return
        x0, t0, n = self._x0, self._t0, self._count
        self._x0 = self._t0 = None
        self._count = 0
        B, K = x0.shape
        if n == 1:
            super().step(x0, t0)
            return
        xall = x0.as_strided((n * B, K), (K, 1))
        tall = t0.as_strided((n * B,) + t0.shape[1:],
                             (t0.stride(0),) + t0.stride()[1:])
        ops.ext().toy_multistep(xall, tall, self.flat_param, self.loss_out,
                                self.use_mse, self.w_off, self.b_off,
                                self.lr, B)


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
        try:
            self._graph = torch.cuda.CUDAGraph()
            with torch.cuda.graph(self._graph):
                super().step(self._x_static, self._t_static)
        except Exception as e:  # capture unsupported -> eager fallback
            print(f"[mi355x_ddp] hipGraph capture failed ({e}); running eager")
            self._graph = False
            torch.cuda.synchronize()

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
