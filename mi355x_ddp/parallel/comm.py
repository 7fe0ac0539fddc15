"""Communicator layer: RCCL-over-xGMI driven directly, with a gloo shadow
for CPU-only testing.

The reference delegates all communication to c10d ProcessGroupNCCL
(init sites multigpu.py:20, multigpu_torchrun.py:13,
multinode_torchrun.py:13, multigpu_profile.py:97). Here torch.distributed
is used ONLY for rendezvous (rank discovery + the TCP store that carries
the 128-byte RCCL unique id); every gradient byte moves
through the framework's own `RcclComm` (ops/csrc/rccl_comm.hip) on a
dedicated comm stream.

`GlooComm` mirrors the same interface over torch.distributed's gloo backend
so the reducer's bucketing/overlap logic is exercised by multi-process CPU
tests (SURVEY §4 consequence (a)).
"""

from __future__ import annotations

import base64
import os
from typing import List, Optional

import torch
import torch.distributed as dist


def ddp_setup(rank: Optional[int] = None, world_size: Optional[int] = None,
              backend: Optional[str] = None) -> None:
    """Process-group bootstrap covering both reference styles:

    - explicit rank/world (mp.spawn style, reference multigpu.py:12-20:
      sets MASTER_ADDR/PORT defaults then init_process_group)
    - env-var style under torchrun (reference multigpu_torchrun.py:12-13).

    The backend defaults to the compound map "cpu:gloo,cuda:nccl" on GPU
    hosts — GPU tensors ride the c10d NCCL(=RCCL) backend exactly as the
    reference's "nccl" does (gradient bytes still go through RcclComm),
    while CPU tensors (the transport-ladder agreement flags and the mesh's
    gloo cross-validation, see all_ranks_agree/P2pMeshComm.validate) ride
    gloo. A bare "nccl" group would reject those CPU collectives. CPU-only
    hosts use "gloo".
    """
    if backend is None:
        backend = "cpu:gloo,cuda:nccl" if torch.cuda.is_available() else "gloo"
    if rank is not None and world_size is not None:
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "12355")
        dist.init_process_group(backend, rank=rank, world_size=world_size)
    elif "TORCHELASTIC_RESTART_COUNT" in os.environ:
        # Restart-safety: under torchrun the agent-hosted TCPStore SURVIVES
        # worker restarts, so plain env:// init can read a crashed peer's
        # STALE transport address from the previous attempt (a race — the
        # new peer may or may not have overwritten the key yet), failing
        # with connection-refused or wedging a rank inside native connect.
        # Keying the process-group store by the elastic attempt number
        # gives every restart a clean namespace.
        attempt = os.environ["TORCHELASTIC_RESTART_COUNT"]
        r = int(os.environ["RANK"])
        w = int(os.environ["WORLD_SIZE"])
        store = dist.TCPStore(os.environ["MASTER_ADDR"],
                              int(os.environ["MASTER_PORT"]),
                              w, is_master=False)
        store = dist.PrefixStore(f"mi355x/attempt_{attempt}", store)
        dist.init_process_group(backend, store=store, rank=r, world_size=w)
    else:
        dist.init_process_group(backend)
    if torch.cuda.is_available():
        if os.environ.get("MI355X_FORCE_DEV0") == "1":
            # test-only: rehearse a multi-rank world on ONE device (IPC
            # time-sharing) — used by the 8-GPU pre-flight on 1-GPU boxes
            torch.cuda.set_device(0)
        else:
            torch.cuda.set_device(int(os.environ.get("LOCAL_RANK",
                                                     rank or 0)))


class GlooComm:
    """CPU shadow / fallback communicator: same interface as the RCCL
    path, backed by torch.distributed gloo. Async handles model the
    comm-stream overlap.

    Collectives go through a DEDICATED gloo group, not the default group:
    under the compound "cpu:gloo,cuda:nccl" backend (ddp_setup), a CUDA
    tensor on the default group routes to ProcessGroupNCCL — which is
    exactly the transport this fallback exists to avoid (e.g. same-device
    ranks, where NCCL raises 'Duplicate GPU detected'). ProcessGroupGloo
    handles CUDA tensors by host staging. All ranks construct GlooComm
    together (the transport ladder agrees first), so the collective
    new_group call is safe."""

    def __init__(self):
        assert dist.is_initialized()
        self.rank = dist.get_rank()
        self.world = dist.get_world_size()
        self._pg = dist.new_group(backend="gloo")
        self._handles: List = []

    def all_reduce_avg(self, t: torch.Tensor) -> None:
        h = dist.all_reduce(t, op=dist.ReduceOp.SUM, group=self._pg,
                            async_op=True)
        self._handles.append((h, t))

    def all_reduce_avg_inline(self, t: torch.Tensor) -> None:
        dist.all_reduce(t, op=dist.ReduceOp.SUM, group=self._pg)
        t.div_(self.world)

    def broadcast(self, t: torch.Tensor, root: int = 0) -> None:
        dist.broadcast(t, src=root, group=self._pg)

    def join_compute(self) -> None:
        for h, t in self._handles:
            h.wait()
            t.div_(self.world)
        self._handles.clear()

    def barrier(self) -> None:
        dist.barrier(group=self._pg)


class RcclCommAdapter:
    """Thin adapter over the native RcclComm (exchange of the unique id via
    the c10d rendezvous store, then direct RCCL)."""

    _KEY = "mi355x_ddp/rccl_uid"

    def __init__(self, device: torch.device):
        from .. import ops
        assert dist.is_initialized()
        self.rank = dist.get_rank()
        self.world = dist.get_world_size()
        store = dist.distributed_c10d._get_default_store()
        if self.rank == 0:
            uid = ops.ext().RcclComm.make_unique_id()
            store.set(self._KEY, base64.b64encode(uid).decode())
        uid = base64.b64decode(store.get(self._KEY))
        self._comm = ops.ext().RcclComm(uid, self.rank, self.world,
                                        device.index or 0)

    def all_reduce_avg(self, t: torch.Tensor) -> None:
        self._comm.all_reduce_avg(t)

    def all_reduce_avg_inline(self, t: torch.Tensor) -> None:
        self._comm.all_reduce_avg_inline(t)

    def broadcast(self, t: torch.Tensor, root: int = 0) -> None:
        self._comm.broadcast(t, root)

    def join_compute(self) -> None:
        self._comm.join_compute()

    def barrier(self) -> None:
        self._comm.barrier()


class P2pMeshComm:
    """Device-side xGMI mesh all-reduce for TINY payloads (the toy step's
    84 B of gradients), layered over a base communicator that keeps
    handling broadcast / barrier / large tensors.

    Setup exchanges HIP IPC mailbox handles through the c10d store, then
    `validate()` cross-checks one mesh all-reduce against the gloo
    process group; any mismatch or kernel timeout raises, so callers can
    fall back to the base (RCCL) transport. The mesh kernel's spin-wait is
    wall-clock-bounded — a lost peer surfaces as a raised error, never a
    hang."""

    _KEY = "mi355x_ddp/p2p_mesh"

    def __init__(self, device: torch.device, base, validate: bool = True):
        from .. import ops
        assert dist.is_initialized()
        self.base = base
        self.rank = dist.get_rank()
        self.world = dist.get_world_size()
        self._mesh = ops.ext().P2pMesh(self.rank, self.world,
                                       device.index or 0)
        store = dist.distributed_c10d._get_default_store()
        store.set(f"{self._KEY}/{self.rank}",
                  base64.b64encode(self._mesh.handle_bytes()).decode())
        handles = []
        for r in range(self.world):
            handles.append(base64.b64decode(store.get(f"{self._KEY}/{r}")))
        self._mesh.connect(handles)
        self._device = device
        if validate:
            self.validate()

    def validate(self) -> None:
        """One mesh all-reduce cross-checked against the c10d group (gloo
        normally; the c10d NCCL backend when the default group is
        nccl-only — still a transport independent of this mesh)."""
        probe = torch.arange(24, dtype=torch.float32, device=self._device)
        probe = probe * (self.rank + 1)
        if _cpu_collectives_ok():
            ref = probe.cpu()
        else:
            ref = probe.clone()  # device tensor for an nccl-only group
        dist.all_reduce(ref, op=dist.ReduceOp.SUM)
        ref = ref.cpu() / self.world
        self._mesh.all_reduce_avg_inline(probe)
        torch.cuda.synchronize()
        self._mesh.check()
        if not torch.allclose(probe.cpu(), ref, atol=1e-5):
            raise RuntimeError(
                "P2pMesh validation mismatch vs gloo all-reduce: "
                f"{(probe.cpu() - ref).abs().max().item()}")

    def check(self) -> None:
        self._mesh.check()

    # payloads at or under this ride the device-side mesh; larger ones the
    # base transport. The threshold is a constant, so routing is identical
    # across ranks WHENEVER the call-site tensor sizes agree — which the
    # reducer guarantees (identical bucket layout on every rank) and
    # MI355X_DEBUG_SYNC=1 verifies per call (all_ranks_same_size).
    MESH_MAX_NUMEL = 64

    def _assert_uniform_size(self, t: torch.Tensor) -> None:
        lohi = torch.tensor([t.numel(), -t.numel()], dtype=torch.int64)
        if not _cpu_collectives_ok():
            lohi = lohi.to(_this_rank_device())
        dist.all_reduce(lohi, op=dist.ReduceOp.MIN)
        lo, hi = int(lohi[0]), -int(lohi[1])
        if lo != hi:
            raise RuntimeError(
                f"P2pMeshComm size-routing divergence: this rank reduces "
                f"{t.numel()} elements but the world spans [{lo}, {hi}] — "
                "ranks would take different transports and hang")

    def all_reduce_avg_inline(self, t: torch.Tensor) -> None:
        if os.environ.get("MI355X_DEBUG_SYNC") == "1":
            self._assert_uniform_size(t)
        if t.numel() <= self.MESH_MAX_NUMEL:
            self._mesh.all_reduce_avg_inline(t)
        else:
            self.base.all_reduce_avg_inline(t)

    # everything else rides the base transport
    def all_reduce_avg(self, t: torch.Tensor) -> None:
        self.base.all_reduce_avg(t)

    def broadcast(self, t: torch.Tensor, root: int = 0) -> None:
        self.base.broadcast(t, root)

    def join_compute(self) -> None:
        self.base.join_compute()

    def barrier(self) -> None:
        self.base.barrier()


def _cpu_collectives_ok() -> bool:
    """Does the default process group accept CPU tensors? Probed from the
    group's per-device backend map (e.g. "cpu:gloo,cuda:nccl") instead of
    retrying on RuntimeError — a transient gloo failure must surface, not
    be misread as 'backend lacks CPU support'. Deterministic across ranks:
    every rank sees the same backend config, so every rank takes the same
    transport branch."""
    try:
        config = str(dist.get_backend_config())
    except Exception:
        config = str(dist.get_backend())
    for entry in config.split(","):
        dev, _, backend = entry.partition(":")
        if not backend:  # bare backend name, applies to all devices
            return dev in ("gloo", "mpi")
        if dev == "cpu":
            return backend in ("gloo", "mpi")
    return False  # no cpu entry: cuda-only (nccl) group


def _this_rank_device() -> torch.device:
    """The CUDA device belonging to THIS local rank. Never bare .cuda():
    for a user-initialized group our ddp_setup (which calls
    torch.cuda.set_device) may not have run, and cuda:0 from every local
    rank would collide in NCCL."""
    if os.environ.get("MI355X_FORCE_DEV0") == "1":
        return torch.device("cuda", 0)
    if "LOCAL_RANK" in os.environ:
        return torch.device("cuda", int(os.environ["LOCAL_RANK"]))
    return torch.device("cuda", torch.cuda.current_device())


def all_ranks_agree(ok: bool) -> bool:
    """World-wide MIN over a per-rank success flag (gloo/default group).

    Per-rank try/except around transport setup is NOT enough: if only
    SOME ranks fall back, the world ends up issuing mismatched collectives
    on different transports, which HANGS instead of failing. Every
    transport decision must therefore be agreed by all ranks — call this
    at the same program point on every rank."""
    t = torch.tensor([1 if ok else 0])
    if not _cpu_collectives_ok():
        # nccl-only default group (user-initialized): agree via a device
        # tensor on THIS rank's device instead.
        t = t.to(_this_rank_device())
    dist.all_reduce(t, op=dist.ReduceOp.MIN)
    return bool(t.item())


def build_gpu_comm(device: torch.device, want_mesh: bool = True,
                   log=print):
    """The full, hang-safe transport ladder for a GPU world:
    RCCL -> (gloo fallback), then optionally the xGMI mesh layered on top
    — with EVERY step agreed across ranks (see all_ranks_agree) and the
    mesh cross-validated against gloo before adoption.
    Returns (comm, kind_string)."""
    assert dist.is_initialized()
    comm = None
    kind = "rccl"
    try:
        comm = RcclCommAdapter(device)
    except Exception as e:
        log(f"[mi355x_ddp] RcclComm init failed on rank "
            f"{dist.get_rank()} ({e!r})")
        kind = "gloo-fallback"
    if not all_ranks_agree(comm is not None):
        comm = GlooComm()
        kind = "gloo-fallback"
    if comm is None:  # this rank failed but (impossible here) others agreed
        comm = GlooComm()
    if want_mesh:
        mesh = None
        try:
            mesh = P2pMeshComm(device, base=comm, validate=False)
        except Exception as e:
            log(f"[mi355x_ddp] P2pMesh setup failed on rank "
                f"{dist.get_rank()} ({e!r})")
        if all_ranks_agree(mesh is not None):
            ok = True
            try:
                mesh.validate()
            except Exception as e:
                log(f"[mi355x_ddp] P2pMesh validation failed on rank "
                    f"{dist.get_rank()} ({e!r})")
                ok = False
            if all_ranks_agree(ok):
                return mesh, "p2p-mesh+" + kind
        log(f"[mi355x_ddp] mesh not adopted; staying on {kind}")
    return comm, kind


def create_comm(device: torch.device):
    """Pick the communicator for this process, or None when world size is 1
    (single-process runs use the same reducer with no collectives).

    GPU worlds go through the SAME hang-safe ladder as bench/the fast
    engines (build_gpu_comm): a rank-asymmetric RcclComm init failure
    downgrades ALL ranks to gloo together instead of leaving the world
    with mismatched transports (the exact hang all_ranks_agree documents).
    The mesh layer is skipped — the generic reducer only issues bucket
    all_reduce_avg calls, which the mesh routes to the base transport
    anyway."""
    if not (dist.is_available() and dist.is_initialized()):
        return None
    if dist.get_world_size() == 1:
        return None
    if device.type == "cuda":
        comm, _kind = build_gpu_comm(device, want_mesh=False)
        return comm
    return GlooComm()
