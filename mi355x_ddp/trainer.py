"""The training engine: a single `Trainer` serving all five entrypoint stages.

The reference keeps four near-identical copies of this class plus a profiler
variant (reference single_gpu.py:6-45, multigpu.py:22-62,
multigpu_torchrun.py:15-68, multinode_torchrun.py:15-69,
multigpu_profile.py:30-91). Here it is one class; the entry scripts configure
it. The observable surface is preserved:

- per-epoch banner `[GPU{n}] Epoch {e} | Batchsize: {b} | Steps: {s}`
  (reference multigpu.py:46-47)
- checkpoint formats: raw state_dict -> checkpoint.pt (single_gpu.py:36-38),
  rank-0 unwrapped state_dict (multigpu.py:53-56,61), snapshot dict
  {"MODEL_STATE", "EPOCHS_RUN"} -> snapshot.pt (multigpu_torchrun.py:57-62)
  restored pre-DDP-wrap (multigpu_torchrun.py:30-34), wrapped state ->
  model_ddp.pth (multigpu_profile.py:76-78)
- resume loop `range(epochs_run, max_epochs)` (multigpu_torchrun.py:64-66)

Reference warts intentionally fixed:
- sampler.set_epoch IS called every epoch;
- no wasted first batch for the banner (batch size read from the loader);
- H2D copies use non_blocking=True with pinned loaders;
- snapshot save guard is GLOBAL rank 0 (the reference's local_rank==0 guard
  races N nodes onto one shared file, multinode_torchrun.py:68);
- torch.load uses map_location onto this rank's device
  (reference omits it: multigpu_torchrun.py:37).
"""

from __future__ import annotations

import os
from typing import Callable, Optional, Union

import torch
from torch.utils.data import DataLoader

from . import ops


def _loss_callable(loss_fn: Union[str, Callable]) -> Callable:
    if callable(loss_fn):
        return loss_fn
    if loss_fn == "ce":
        return ops.cross_entropy
    if loss_fn == "mse":
        return ops.mse_loss
    raise ValueError(f"unknown loss_fn {loss_fn!r}")


class Trainer:
    """Epoch/batch training loop with periodic checkpointing and optional
    snapshot-based fault-tolerant resume.

    Args mirror the reference Trainer family; extras are keyword-only.

    gpu_id: device index, or "cpu" for the CPU plumbing path, or None to
        read LOCAL_RANK from the torchrun env (multigpu_torchrun.py:24).
    snapshot_path: when set, enables the fault-tolerance contract of
        reference stage 3/4 (load-if-exists before DDP wrap, save snapshot
        every save_every epochs on global rank 0).
    wrap_ddp: wrap the model in mi355x_ddp.parallel.DDP when a process
        group is initialized (reference wraps unconditionally in the
        distributed scripts, multigpu.py:36).
    """

    def __init__(
        self,
        model: torch.nn.Module,
        train_data: DataLoader,
        optimizer: torch.optim.Optimizer,
        gpu_id: Optional[Union[int, str]] = None,
        save_every: int = 1,
        *,
        snapshot_path: Optional[str] = None,
        checkpoint_path: str = "checkpoint.pt",
        loss_fn: Union[str, Callable] = "ce",
        wrap_ddp: bool = True,
        profile: bool = False,
        profile_dir: str = "./log/resnet50/",
        save_wrapped: bool = False,
        bucket_cap_mb: Optional[float] = None,
        engine: str = "hooks",
    ) -> None:
        """engine: "hooks" (default — the generic autograd+reducer path,
        reference semantics for arbitrary models), "hooks-graph" (the same
        generic path captured in one hipGraph and replayed — fast for any
        static-shape model; falls back to eager hooks if capture fails or
        off-GPU; NOTE: capture warmup performs a few extra real training
        steps on the first batch, per torch's capture recipe), or "fused"/"persistent"/"graph"/"auto" to run the toy
        fast path (single-launch fused step / multi-step deferred kernel /
        hipGraph replay) when the model+loss+optimizer qualify; silently
        falls back to hooks otherwise. Unknown values raise."""
        if engine not in ("hooks", "hooks-graph", "auto", "fused",
                          "persistent", "graph"):
            raise ValueError(f"unknown engine {engine!r} (hooks | "
                             "hooks-graph | auto | fused | persistent | "
                             "graph)")
        if gpu_id is None:
            gpu_id = int(os.environ.get("LOCAL_RANK", 0))
        if gpu_id != "cpu" and os.environ.get("MI355X_FORCE_DEV0") == "1":
            gpu_id = 0  # test-only: multi-rank world on ONE device
        self.gpu_id = gpu_id
        self.device = torch.device("cpu") if gpu_id == "cpu" else torch.device("cuda", gpu_id)
        # global rank: the process group is authoritative (mp.spawn sets no
        # RANK env; torchrun does — reference multinode_torchrun.py:25)
        import torch.distributed as dist
        if dist.is_available() and dist.is_initialized():
            self.global_rank = dist.get_rank()
        else:
            self.global_rank = int(os.environ.get("RANK", 0))
        self.train_data = train_data
        self.optimizer = optimizer
        self.save_every = save_every
        self.snapshot_path = snapshot_path
        self.checkpoint_path = checkpoint_path
        self.loss_fn = _loss_callable(loss_fn)
        self.save_wrapped = save_wrapped
        self.epochs_run = 0
        self.profile = profile
        self.profile_dir = profile_dir
        # copy-stream H2D prefetch (on by default on GPU; MI355X_PREFETCH=0
        # falls back to inline non_blocking copies)
        self._use_prefetcher = (self.device.type == "cuda"
                                and os.environ.get("MI355X_PREFETCH", "1") != "0")

        self.model = model.to(self.device)
        # Measured-config knobs (profiles/README.md r02d: bf16-autocast +
        # NHWC is the fastest ResNet-50 configuration, 12.95 ms/step vs
        # 17.5 fp32 NCHW): MI355X_NHWC=1 converts the model to
        # channels_last (MIOpen's preferred layout; inputs converted per
        # batch), MI355X_AUTOCAST_BF16=1 runs forward+loss under bf16
        # autocast (losses stay fp32 per the ops autocast policy).
        self._nhwc = (self.device.type == "cuda"
                      and os.environ.get("MI355X_NHWC") == "1")
        self._autocast = (self.device.type == "cuda"
                          and os.environ.get("MI355X_AUTOCAST_BF16") == "1")
        if self._nhwc:
            self.model = self.model.to(memory_format=torch.channels_last)

        # Restore BEFORE any DDP wrap, as the reference does
        # (multigpu_torchrun.py:30-34): ranks then start from identical
        # weights and the wrap-time broadcast is a no-op check.
        if snapshot_path and os.path.exists(snapshot_path):
            print("Loading snapshot")
            self._load_snapshot(snapshot_path)

        self._distributed = False
        self._engine = None
        if engine not in ("hooks", "hooks-graph") \
                and self._try_fast_engine(engine):
            pass  # _run_batch drives self._engine
        elif wrap_ddp and dist.is_available() and dist.is_initialized():
            # world 1 wraps too: the engine still provides flat buckets and
            # the fused SGD path (there is simply no communicator).
            from .parallel import DDP, FusedSGD
            graphed = engine == "hooks-graph" and self.device.type == "cuda"
            self.model = DDP(self.model, bucket_cap_mb=bucket_cap_mb,
                             cpp_hooks=False if graphed else None)
            self._distributed = True
            if isinstance(self.optimizer, FusedSGD):
                self.optimizer.attach_reducer(self.model.reducer)
            # A bucket-attached FusedSGD zeroes the flat grads inside its
            # step kernel; any other optimizer needs an explicit zero or
            # gradients would accumulate across steps.
            self._needs_zero = not (
                isinstance(self.optimizer, FusedSGD)
                and self.optimizer._flat_pairs is not None)
            if engine == "hooks-graph" and self.device.type == "cuda" \
                    and not (self._autocast or self._nhwc):
                # (the autocast/NHWC knobs run the plain loop — the
                # graphed engine's captured step doesn't re-apply them)
                # whole-step hipGraph capture of the generic path: one
                # replay per step instead of ~10 launches + autograd
                # overhead. Eager-fallback (with a warning) on capture
                # failure — training results are identical either way.
                from .engine import GraphedAutogradStep
                self._engine = GraphedAutogradStep(
                    self.model, self.loss_fn, self.optimizer,
                    finalize=self.model.finalize_backward,
                    zero_grad=self._needs_zero)

    def _try_fast_engine(self, kind: str) -> bool:
        """Engage the toy fast path when the configuration qualifies:
        HipLinear(K,1) model, MSE loss, FusedSGD with a single lr, CUDA
        device. Returns False (caller keeps the hooks path) otherwise."""
        import torch.distributed as dist
        from .models.toy import HipLinear
        from .parallel import FusedSGD
        m = self.model
        if not (isinstance(m, HipLinear) and m.out_features == 1
                and m.bias is not None
                and self.device.type == "cuda"
                and self.loss_fn is ops.mse_loss
                and isinstance(self.optimizer, FusedSGD)
                and len(self.optimizer.param_groups) == 1):
            return False
        if dist.is_available() and dist.is_initialized() \
                and dist.get_world_size() > 1:
            # hang-safe, world-agreed transport ladder (RCCL -> gloo, +
            # mesh only when every rank set it up AND it cross-validated)
            from .parallel.comm import build_gpu_comm
            comm, _kind = build_gpu_comm(self.device)
        else:
            comm = None
        if kind == "auto":
            kind = "persistent"
        from .engine import GraphedToyStep, PersistentToyStep, ToyFusedStep
        lr = self.optimizer.param_groups[0]["lr"]
        if kind == "persistent":
            if comm is not None and getattr(comm, "_mesh", None) is None:
                kind = "fused"  # multi-step needs world 1 or a mesh comm
        cls = {"persistent": PersistentToyStep, "graph": GraphedToyStep,
               "fused": ToyFusedStep}[kind]
        self._engine = cls(m, comm=comm, lr=lr, use_mse=True)
        if comm is not None:
            self._engine.reducer.broadcast_params(root=0)
            self._distributed = True
        return True

    # -- snapshot / checkpoint -------------------------------------------
    def _unwrapped(self) -> torch.nn.Module:
        return self.model.module if hasattr(self.model, "module") else self.model

    def _load_snapshot(self, snapshot_path: str) -> None:
        # Byte-compatible with the reference snapshot (torch.save zip+pickle,
        # keys MODEL_STATE / EPOCHS_RUN, multigpu_torchrun.py:57-62).
        snapshot = torch.load(snapshot_path, map_location=self.device, weights_only=True)
        self._unwrapped().load_state_dict(snapshot["MODEL_STATE"])
        self.epochs_run = snapshot["EPOCHS_RUN"]
        print(f"Resuming training from snapshot at Epoch {self.epochs_run}")

    @staticmethod
    def _atomic_save(obj, path: str) -> None:
        """torch.save via tmp + os.replace. The reference writes the live
        file in place (multigpu_torchrun.py:61): a worker crash mid-save
        corrupts snapshot.pt, and every elastic restart then dies loading
        it — a permanent crash loop. Rename is atomic on POSIX, so readers
        (including other nodes on a shared mount, SURVEY §2.1 'multinode
        snapshot race') see either the old or the new complete file."""
        tmp = f"{path}.tmp.{os.getpid()}"
        try:
            torch.save(obj, tmp)
            os.replace(tmp, path)
        except BaseException:
            try:
                os.remove(tmp)
            except OSError:
                pass
            raise

    def _save_snapshot(self, epoch: int) -> None:
        snapshot = {
            "MODEL_STATE": self._unwrapped().state_dict(),
            "EPOCHS_RUN": epoch,
        }
        self._atomic_save(snapshot, self.snapshot_path)
        print(f"Epoch {epoch} | Training snapshot saved at {self.snapshot_path}")

    def _save_checkpoint(self, epoch: int) -> None:
        if self.save_wrapped:
            ckp = self.model.state_dict()  # DDP-prefixed, multigpu_profile.py:76-78
        else:
            ckp = self._unwrapped().state_dict()
        self._atomic_save(ckp, self.checkpoint_path)
        print(f"Epoch {epoch} | Training checkpoint saved at {self.checkpoint_path}")

    # -- the hot loop -----------------------------------------------------
    def _run_batch(self, source: torch.Tensor, targets: torch.Tensor) -> None:
        # fwd -> loss -> bwd (+ overlapped bucket all-reduce) -> step.
        # Reference single_gpu.py:21-26. zero_grad is folded into the
        # bucket lifecycle when the model is DDP-wrapped (the fused SGD
        # zeroes the flat grad buffer); otherwise we zero here.
        if self._engine is not None:
            self._engine.step(source.contiguous(), targets.contiguous())
            return
        if not self._distributed or getattr(self, "_needs_zero", False):
            self.optimizer.zero_grad(set_to_none=False)
        if self._nhwc and source.dim() == 4:
            source = source.contiguous(memory_format=torch.channels_last)
        if self._autocast:
            with torch.autocast("cuda", dtype=torch.bfloat16):
                output = self.model(source)
                loss = self.loss_fn(output, targets)
        else:
            output = self.model(source)
            loss = self.loss_fn(output, targets)
        loss.backward()
        if self._distributed:
            self.model.finalize_backward()
        self.optimizer.step()

    def _maybe_inject_fault(self, epoch: int) -> None:
        """Fault injection for elastic-restart testing (the
        reference has no injection tooling; resume-by-kill is its implied
        validation). When MI355X_FAULT_EPOCH matches and the one-shot marker
        file does not exist yet, this rank hard-exits — torchrun's elastic
        agent then restarts the job, which must resume from the snapshot."""
        if os.environ.get("MI355X_FAULT_EPOCH") != str(epoch):
            return
        if self.global_rank != int(os.environ.get("MI355X_FAULT_RANK", "0")):
            return
        marker = os.environ.get("MI355X_FAULT_ONCE_FILE")
        if marker:
            if os.path.exists(marker):
                return  # already crashed once; run through this time
            with open(marker, "w") as f:
                f.write("crashed\n")
        print(f"[GPU{self.global_rank}] injected fault at epoch {epoch}",
              flush=True)
        os._exit(17)

    def _run_epoch(self, epoch: int) -> None:
        self._maybe_inject_fault(epoch)
        b_sz = self.train_data.batch_size
        print(f"[GPU{self.global_rank if self._distributed else self.gpu_id}] "
              f"Epoch {epoch} | Batchsize: {b_sz} | Steps: {len(self.train_data)}")
        sampler = getattr(self.train_data, "sampler", None)
        if sampler is not None and hasattr(sampler, "set_epoch"):
            sampler.set_epoch(epoch)
        if self._use_prefetcher:
            # copy-stream pipeline: batches arrive already on device,
            # uploaded one step ahead (fixes the reference's
            # synchronous .to(), multigpu.py:49-50)
            from .data import DevicePrefetcher
            batches = DevicePrefetcher(self.train_data, self.device)
        else:
            batches = self.train_data
        non_blocking = self.device.type == "cuda"
        for source, targets in batches:
            if source.device != self.device:
                source = source.to(self.device, non_blocking=non_blocking)
                targets = targets.to(self.device, non_blocking=non_blocking)
            self._run_batch(source, targets)
            if self._profiler is not None:
                self._profiler.step()
        if self._engine is not None and hasattr(self._engine, "flush"):
            self._engine.flush()  # deferred engines: run any pending steps

    def _create_profiler(self):
        # Parity with reference multigpu_profile.py:80-91: schedule
        # wait=1/warmup=1/active=5, CPU+GPU activities, TensorBoard trace
        # per rank. Kineto is roctracer-backed on ROCm. Beyond parity
        # (VERDICT r01 item 6): each trace window ALSO emits a per-rank
        # kernel summary (rocprofv3-style top-kernels table + JSON) into
        # profile_dir, so the profile stage produces its evidence in one
        # command at any world size.
        from torch.profiler import (ProfilerActivity, profile,
                                    schedule, tensorboard_trace_handler)
        activities = [ProfilerActivity.CPU]
        if self.device.type == "cuda":
            activities.append(ProfilerActivity.CUDA)
        tb = tensorboard_trace_handler(self.profile_dir,
                                       worker_name=str(self.global_rank))

        def on_ready(prof):
            tb(prof)
            self._write_kernel_summary(prof)

        return profile(
            schedule=schedule(wait=1, warmup=1, active=5),
            activities=activities,
            on_trace_ready=on_ready,
        )

    def _write_kernel_summary(self, prof) -> None:
        """Per-rank kernel stats from the profiled window: a human table
        and a machine-readable JSON (name, calls, device time)."""
        import json
        ka = prof.key_averages()
        rows = []
        for ev in ka:
            dev_us = getattr(ev, "self_device_time_total", 0) or 0
            if dev_us <= 0:
                continue
            rows.append({"name": ev.key, "calls": ev.count,
                         "device_time_us": dev_us})
        rows.sort(key=lambda r: -r["device_time_us"])
        base = os.path.join(self.profile_dir,
                            f"kernel_stats_rank{self.global_rank}")
        os.makedirs(self.profile_dir, exist_ok=True)
        with open(base + ".json", "w") as f:
            json.dump(rows, f, indent=1)
        with open(base + ".txt", "w") as f:
            f.write(ka.table(sort_by="self_cuda_time_total"
                             if self.device.type == "cuda"
                             else "self_cpu_time_total", row_limit=25))
        print(f"[GPU{self.global_rank}] kernel summary -> {base}.json "
              f"({len(rows)} device kernels)", flush=True)

    def train(self, max_epochs: int) -> None:
        self._profiler = self._create_profiler() if self.profile else None
        if self._profiler is not None:
            self._profiler.start()
        try:
            for epoch in range(self.epochs_run, max_epochs):
                self._run_epoch(epoch)
                if self.global_rank == 0 and epoch % self.save_every == 0:
                    if self.snapshot_path:
                        self._save_snapshot(epoch)
                    else:
                        self._save_checkpoint(epoch)
        finally:
            if self._profiler is not None:
                self._profiler.stop()

    # reference multigpu_profile.py exposes run_epoch/save_checkpoint
    def run_epoch(self, nb_epochs: int) -> None:
        self.train(nb_epochs)

    def save_checkpoint(self) -> None:
        self._save_checkpoint(self.epochs_run)
