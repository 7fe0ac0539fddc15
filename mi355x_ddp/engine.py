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
