"""DistributedDataParallel replacement built on the native Reducer + RCCL.

API parity with the reference's DDP usage:
- `DDP(model)` at wrap time broadcasts parameters and buffers from rank 0
  (reference wrap sites multigpu.py:36, multigpu_torchrun.py:34,
  multinode_torchrun.py:35, multigpu_profile.py:41; broadcast = N4)
- `.module` exposes the wrapped model for checkpoint unwrapping
  (reference multigpu.py:54 `model.module.state_dict()`)
- the per-step contract is loss.backward() -> finalize_backward() ->
  optimizer.step() (the Trainer calls finalize_backward; stock DDP hides
  the equivalent fence inside its autograd hooks)

Also usable with world_size == 1 (no communicator): the model still gets
flat parameter/grad buckets and the fused SGD path, which is how the
single-GPU benchmark runs the same engine.
"""

from __future__ import annotations

from contextlib import contextmanager

from typing import Optional

import torch

from .comm import create_comm
from .reducer import Reducer


class DDP(torch.nn.Module):
    def __init__(self, module: torch.nn.Module,
                 device_ids=None, output_device=None,
                 bucket_cap_mb: Optional[float] = None,
                 comm=None, grad_views: Optional[bool] = None,
                 broadcast_buffers: bool = True,
                 cpp_hooks: Optional[bool] = None,
                 find_unused_parameters: bool = False,
                 gradient_as_bucket_view: Optional[bool] = None,
                 static_graph: bool = False):
        # Stock-DDP drop-in kwargs: find_unused_parameters is a no-op —
        # models with conditionally-unused params train here BY DEFAULT
        # (finalize() launches straggler buckets; stock hangs without the
        # flag). gradient_as_bucket_view maps onto grad_views (views are
        # already our default). static_graph is accepted and ignored (an
        # optimization hint for stock's reducer rebuild).
        if gradient_as_bucket_view is not None and grad_views is None:
            grad_views = gradient_as_bucket_view
        del find_unused_parameters, static_graph
        # device_ids/output_device: accepted for drop-in compatibility with
        # the stock signature the reference uses (`DDP(model,
        # device_ids=[gpu_id])`, ref multigpu.py:36). The model must already
        # live on that device (our Trainer moves it first, as the reference
        # does); a mismatch is an error, not a silent migration.
        super().__init__()
        if device_ids:
            if len(device_ids) != 1:
                raise ValueError("single-device module expected (like stock "
                                 "DDP's single-process-single-device mode)")
            want = torch.device("cuda", device_ids[0]) \
                if isinstance(device_ids[0], int) else torch.device(device_ids[0])
            if want.type == "cuda" and want.index is None:
                # indexless "cuda" (stock DDP accepts it): means the current
                # device, like tensor.cuda() would
                want = torch.device("cuda", torch.cuda.current_device())
            for name, p in module.named_parameters():
                have = p.device
                if have.type == "cuda" and have.index is None:
                    have = torch.device("cuda", torch.cuda.current_device())
                if (have.type, have.index) != (want.type, want.index):
                    raise ValueError(
                        f"parameter {name} is on {p.device} but device_ids "
                        f"names {want}; move the model first (model.to(device))")
        self.module = module
        params = list(module.parameters())
        device = params[0].device if params else torch.device("cpu")
        self.comm = comm if comm is not None else create_comm(device)
        self.reducer = Reducer(params, comm=self.comm,
                               bucket_cap_mb=bucket_cap_mb,
                               grad_views=grad_views,
                               cpp_hooks=cpp_hooks)
        # wrap-time module-state sync: params coalesced per bucket,
        # buffers (e.g. BN running stats) coalesced per dtype (SURVEY §2.4
        # row 2). With broadcast_buffers=True (stock DDP's default) the
        # buffer sync ALSO runs before every forward, so BN running stats
        # never drift across ranks.
        self.broadcast_buffers = broadcast_buffers
        self._buf_groups = []
        if self.comm is not None:
            groups = {}
            for buf in module.buffers():
                if buf.numel():
                    groups.setdefault(buf.dtype, []).append(buf)
            for bufs in groups.values():
                flat = torch.empty(sum(b.numel() for b in bufs),
                                   dtype=bufs[0].dtype, device=bufs[0].device)
                self._buf_groups.append((flat, bufs))
            self.reducer.broadcast_params(root=0)
            self._sync_buffers()

    def _sync_buffers(self) -> None:
        """Rank-0 -> all broadcast of module buffers, one collective per
        dtype group (a ResNet-50 has ~106 BN buffers; coalescing keeps
        this at 2 collectives instead of 106)."""
        with torch.no_grad():
            for flat, bufs in self._buf_groups:
                off = 0
                for b in bufs:
                    flat[off:off + b.numel()].copy_(b.reshape(-1))
                    off += b.numel()
                self.comm.broadcast(flat, 0)
                off = 0
                for b in bufs:
                    b.copy_(flat[off:off + b.numel()].view_as(b))
                    off += b.numel()

    def forward(self, *args, **kwargs):
        if self.broadcast_buffers and self._buf_groups:
            self._sync_buffers()
        return self.module(*args, **kwargs)

    def finalize_backward(self) -> None:
        """Launch straggler buckets and fence compute on the comm stream.
        Must be called between loss.backward() and optimizer.step()."""
        self.reducer.finalize()

    @contextmanager
    def no_sync(self):
        """Gradient accumulation, stock-DDP style (beyond reference parity):
        backward passes inside the context skip the bucket all-reduces;
        gradients accumulate locally (flat-bucket views, or p.grad in copy
        mode) and the first backward OUTSIDE the context reduces the
        accumulated totals. Do not call optimizer.step() inside the
        context, and call finalize_backward() after every backward (it
        resets the per-step ready counters either way)."""
        self.reducer.skip_comm = True
        try:
            yield
        finally:
            self.reducer.skip_comm = False

    # parity helper: stock DDP state_dict carries the "module." prefix;
    # nn.Module gives us that for free since `module` is a submodule.
