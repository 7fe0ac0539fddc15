"""The gradient-bucket Reducer: the centerpiece of the native DDP engine
(SURVEY §2.2 N3 — torch's C++ `reducer.cpp` equivalent, re-designed for
MI355X).

Design (vs. the stock NVSwitch-tuned reducer):

- Buckets are assigned in REVERSE registration order (gradients become
  ready roughly in that order during backward), size-capped by
  MI355X_BUCKET_MB (default 25 MB; first bucket MI355X_FIRST_BUCKET_MB,
  default 1 MB, so the first all-reduce fires early). On an 8-GPU MI355X
  node the collectives ride 7 point-to-point xGMI links per GPU
  (SURVEY §5.8); the cap is an env knob precisely so it can be re-derived
  per topology instead of hard-coding the NVSwitch-era default.

- Both the parameters and the gradients of a bucket live in flat,
  4-element-aligned device buffers. Parameters are REBOUND as views into
  flat_param at construction: the wrap-time rank-0 broadcast
  is one collective per bucket, and the fused SGD step is one kernel per
  bucket over the flat pair.

- Gradient transport has two modes:
  * views (default, MI355X_GRAD_VIEWS=1): p.grad is a view into flat_grad,
    so autograd accumulates straight into the bucket — zero-copy; the
    post-accumulate hook only counts readiness.
  * copy (MI355X_GRAD_VIEWS=0): autograd owns stable grad tensors; a
    bucket-ready event triggers ONE flatten kernel launch (gather all
    member grads into the bucket and zero the sources — the reference's
    implicit flatten + zero_grad, SURVEY N3+N9).

- When a bucket is ready its all-reduce(avg) is launched on the dedicated
  comm stream — but strictly in bucket order across ranks (collectives
  must be issued in identical order on every rank). `finalize()` launches
  any straggler buckets (unused params contribute zeros), then fences the
  compute stream on the comm stream (SURVEY §3.5 'finalize_backward').
"""

from __future__ import annotations

import os
from typing import List, Optional

import torch

from .. import ops


def _env_mb(name: str, default: float) -> float:
    try:
        return float(os.environ.get(name, default))
    except ValueError:
        return default


class Bucket:
    __slots__ = ("index", "params", "offsets", "numel", "flat_param",
                 "flat_grad", "pending", "ready", "plan", "nblocks",
                 "plan_grads", "plan_params")

    def __init__(self, index: int, params: List[torch.nn.Parameter],
                 offsets: List[int], numel: int, device: torch.device,
                 dtype: torch.dtype):
        self.index = index
        self.params = params
        self.offsets = offsets
        self.numel = numel
        self.flat_param = torch.zeros(numel, device=device, dtype=dtype)
        self.flat_grad = torch.zeros(numel, device=device, dtype=dtype)
        self.pending = len(params)
        self.ready = False
        self.plan = None      # device copy plan for the flatten kernel
        self.nblocks = 0
        self.plan_grads = None  # keeps the planned grad tensors alive
        self.plan_params = None  # members included in the plan, in order

    def grad_view(self, i: int) -> torch.Tensor:
        p = self.params[i]
        off = self.offsets[i]
        return self.flat_grad[off:off + p.numel()].view_as(p)


class Reducer:
    def __init__(self, params: List[torch.nn.Parameter], comm=None,
                 bucket_cap_mb: Optional[float] = None,
                 grad_views: Optional[bool] = None,
                 cpp_hooks: Optional[bool] = None):
        params = [p for p in params if p.requires_grad]
        if not params:
            raise ValueError("Reducer needs at least one trainable parameter")
        self.comm = comm
        device = params[0].device
        dtype = params[0].dtype
        for p in params:
            if p.device != device or p.dtype != dtype:
                raise ValueError("Reducer v1 requires uniform device/dtype")
        if grad_views is None:
            grad_views = os.environ.get("MI355X_GRAD_VIEWS", "1") != "0"
        self.grad_views = grad_views

        cap = bucket_cap_mb if bucket_cap_mb is not None else _env_mb("MI355X_BUCKET_MB", 25.0)
        first_cap = min(cap, _env_mb("MI355X_FIRST_BUCKET_MB", 1.0))
        elem = dtype.itemsize
        cap_elems = max(1, int(cap * 1024 * 1024 / elem))
        first_cap_elems = max(1, int(first_cap * 1024 * 1024 / elem))

        # reverse registration order: grads become ready ~in this order
        order = list(reversed(params))
        self.buckets: List[Bucket] = []
        cur: List[torch.nn.Parameter] = []
        cur_offsets: List[int] = []
        cur_numel = 0

        def close_bucket():
            nonlocal cur, cur_offsets, cur_numel
            if not cur:
                return
            padded = (cur_numel + 3) & ~3
            b = Bucket(len(self.buckets), cur, cur_offsets, max(padded, 4),
                       device, dtype)
            self.buckets.append(b)
            cur, cur_offsets, cur_numel = [], [], 0

        for p in order:
            n = p.numel()
            limit = first_cap_elems if not self.buckets else cap_elems
            if cur and cur_numel + n > limit:
                close_bucket()
            cur_offsets.append(cur_numel)
            cur.append(p)
            cur_numel += (n + 3) & ~3  # 4-elem alignment for float4 paths
        close_bucket()

        # Move parameters into the flat buffers and (in views mode) point
        # p.grad at the bucket.
        self._param_index = {}
        for b in self.buckets:
            for i, p in enumerate(b.params):
                off = b.offsets[i]
                pview = b.flat_param[off:off + p.numel()].view_as(p)
                with torch.no_grad():
                    pview.copy_(p.data)
                p.data = pview
                if self.grad_views:
                    p.grad = b.grad_view(i)
                self._param_index[p] = (b, i)
                p._mi355x_bucket = b.index  # used by FusedSGD

        # Hook transport: the C++ ReducerCore installs post-hooks directly
        # on the AccumulateGrad nodes — ready-counting and the bucket
        # all-reduce launch run on the autograd engine thread without the
        # GIL (torch's own reducer.cpp is C++ for the same
        # reason). Engaged on the GPU views path whenever the communicator
        # is native RCCL (or absent); the Python hook path remains for
        # CPU/gloo testing and copy-mode. MI355X_CPP_HOOKS=0 forces the
        # Python path (used by the GPU parity tests).
        self._core = None
        self._hooks = []
        self._skip_comm = False  # set by DDP.no_sync() during accumulation
        self._unfenced = False   # collectives launched but not yet fenced
        if cpp_hooks is None:
            cpp_hooks = os.environ.get("MI355X_CPP_HOOKS", "1") != "0"
        # cpp_hooks=False is REQUIRED under whole-step hipGraph capture
        # (GraphedAutogradStep): a C++ node post-hook — even a no-op one —
        # segfaults hipStreamEndCapture on this torch/ROCm build (bisected
        # on hardware, profiles/README.md r02). Hooks never execute during
        # replay, so the captured path loses nothing.
        use_core = (self.grad_views and device.type == "cuda"
                    and cpp_hooks and ops.has_ext())
        core_comm = None
        if use_core and comm is not None:
            from .comm import P2pMeshComm, RcclCommAdapter
            base = comm.base if isinstance(comm, P2pMeshComm) else comm
            if isinstance(base, RcclCommAdapter):
                core_comm = base._comm
            else:
                use_core = False  # gloo shadow: Python hooks mirror it
        if use_core:
            self._core = ops.ext().ReducerCore(
                [list(b.params) for b in self.buckets],
                [[b.grad_view(i) for i in range(len(b.params))]
                 for b in self.buckets],
                [b.flat_grad for b in self.buckets],
                core_comm)
            self._core.attach_hooks()
        else:
            self._hooks = [
                p.register_post_accumulate_grad_hook(self._make_hook(p))
                for b in self.buckets for p in b.params
            ]
        self._next_launch = 0

    @property
    def skip_comm(self) -> bool:
        return self._skip_comm

    @skip_comm.setter
    def skip_comm(self, v: bool) -> None:
        self._skip_comm = v
        if self._core is not None:
            self._core.set_skip_comm(v)

    # -- wrap-time state sync --------------------------------
    def broadcast_params(self, root: int = 0) -> None:
        if self.comm is None:
            return
        for b in self.buckets:
            self.comm.broadcast(b.flat_param, root)

    # -- backward-side machinery ----------------------------------------
    def _make_hook(self, p: torch.nn.Parameter):
        # bucket/index resolved at registration time (_param_index is
        # final by then) — the hook body runs once per param per step
        b, idx = self._param_index[p]
        grad_views = self.grad_views  # fixed for the Reducer's lifetime

        def hook(param: torch.nn.Parameter) -> None:
            if grad_views:
                g = param.grad
                view = b.grad_view(idx)
                if g is not None and g.data_ptr() != view.data_ptr():
                    # autograd replaced our view (first iteration, or an
                    # out-of-place accumulation): fold into the bucket and
                    # re-bind so the next backward accumulates in place.
                    with torch.no_grad():
                        view.add_(g)
                    param.grad = view
            b.pending -= 1
            if b.pending == 0:
                b.ready = True
                self._launch_ready_in_order()
        return hook

    @property
    def unfenced(self) -> bool:
        """True while bucket collectives are launched but not yet fenced
        by finalize() — reading flat_grad in this window is the reducer's
        one real race. Checked by FusedSGD under
        MI355X_DEBUG_SYNC=1."""
        if self._core is not None:
            return self._core.unfenced
        return self._unfenced

    def _launch_ready_in_order(self) -> None:
        while (self._next_launch < len(self.buckets)
               and self.buckets[self._next_launch].ready):
            b = self.buckets[self._next_launch]
            if not self.grad_views and not self.skip_comm:
                # during no_sync, leave grads accumulating in p.grad —
                # flatten zeroes its sources, which would drop them
                self._flatten_bucket(b)
            if self.comm is not None and not self.skip_comm:
                self.comm.all_reduce_avg(b.flat_grad)
                self._unfenced = True
            self._next_launch += 1

    def _flatten_bucket(self, b: Bucket) -> None:
        if b.flat_grad.is_cuda:
            # The copy plan captures raw grad POINTERS, so it is only valid
            # while every member grad is still the same tensor autograd
            # accumulates into. If any was replaced (p.grad = None between
            # steps, out-of-place accumulation) or a previously-unused
            # member now has a grad, rebuild — a stale plan would silently
            # flatten last iteration's memory or drop the new gradient.
            if b.plan is not None and (
                    any(p.grad is not g
                        for p, g in zip(b.plan_params, b.plan_grads))
                    or sum(p.grad is not None for p in b.params)
                    != len(b.plan_params)):
                b.plan = None
            if b.plan is None:
                grads, offs, members = [], [], []
                for i, p in enumerate(b.params):
                    if p.grad is None:
                        continue  # unused param: its segment stays zero
                    assert p.grad.is_contiguous(), "grad must be contiguous"
                    grads.append(p.grad)
                    offs.append(b.offsets[i])
                    members.append(p)
                b.plan = ops.ext().build_copy_plan(
                    grads, offs, b.flat_grad.device.index or 0)
                b.nblocks = b.plan.shape[0]
                b.plan_grads = grads  # plan holds raw pointers: keep alive
                b.plan_params = members
            # one launch: gather member grads into the bucket AND zero the
            # sources (the reference's implicit flatten + zero_grad)
            ops.ext().flatten_into(b.flat_grad, b.plan, b.nblocks, True)
        else:
            with torch.no_grad():
                for i, p in enumerate(b.params):
                    if p.grad is not None:
                        b.grad_view(i).copy_(p.grad)
                        p.grad.zero_()

    def finalize(self) -> None:
        """Called after loss.backward(): launch stragglers, fence compute on
        the comm stream, and reset per-step state."""
        if self._core is not None:
            self._core.finalize()  # C++: stragglers + fence + reset, no GIL
            return
        for b in self.buckets[self._next_launch:]:
            if not self.grad_views and not self.skip_comm:
                self._flatten_bucket(b)
            if self.comm is not None and not self.skip_comm:
                self.comm.all_reduce_avg(b.flat_grad)
        self._next_launch = len(self.buckets)
        if self.comm is not None and not self.skip_comm:
            self.comm.join_compute()
        self._unfenced = False
        for b in self.buckets:
            b.pending = len(b.params)
            b.ready = False
        self._next_launch = 0

    # -- introspection ----------------------------------------------------
    def flat_pairs(self):
        return [(b.flat_param, b.flat_grad) for b in self.buckets]

    def detach_hooks(self) -> None:
        if self._core is not None:
            self._core.detach_hooks()
        for h in self._hooks:
            h.remove()
