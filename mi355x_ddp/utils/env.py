"""Env-var knobs (the reference's only config channels are positional argv
and env vars — SURVEY §5.6; internal engine knobs live here).

MI355X_BUCKET_MB        gradient bucket cap (default 25)
MI355X_FIRST_BUCKET_MB  first-bucket cap so the first all-reduce fires
                        early (default 1)
MI355X_GRAD_VIEWS       1 = grads are views into flat buckets (zero-copy,
                        default); 0 = flatten-kernel gather per bucket
MI355X_SHUFFLE          'randperm' = torch.randperm epoch shuffle in
                        bench.py's device data path (default: fused
                        in-kernel bijection, ops.perm_index)
MI355X_EPOCH_SIDE_STREAM  1 = gather epoch shards on a side stream
                        (measured slower; default 0)
MI355X_PREFETCH         0 = disable the Trainer's copy-stream H2D
                        prefetcher (default on for GPU)
MI355X_P2P              0 = disable the xGMI mesh all-reduce in bench.py
                        (default on; mesh is only adopted after all ranks
                        agree it set up and cross-validated)
MI355X_EPOCH_BLOCK      epochs gathered per epoch_shard_multi launch in
                        bench.py, letting the multistep launch span epoch
                        boundaries (default: fills the engine's max_defer
                        window; 1 = per-epoch gathers)
MI355X_ENGINE           fast-engine selection for the entrypoint scripts
                        (hooks default | hooks-graph | auto | fused |
                        persistent | graph; silently falls back to hooks
                        when the stage does not qualify)
MI355X_DTYPE            bf16 = cast the model in multigpu.py (BASELINE
                        config 2); unset = the reference's fp32
MI355X_CPP_HOOKS        0 = Python reducer hooks instead of the C++
                        ReducerCore on the GPU views path (default 1;
                        graph capture forces the Python path internally —
                        C++ node post-hooks segfault capture_end)
MI355X_DEBUG_SYNC       1 = stream-race assertions:
                        FusedSGD refuses unfenced bucket reads; the mesh
                        verifies size routing across ranks per call
MI355X_COMM_PRIO        comm-stream priority: unset/0 = DEFAULT priority
                        (required — any elevated priority on a live
                        communicator corrupts MIOpen's conv solution
                        search, profiles r02c); <int>|greatest for
                        experiments only
MI355X_AUTOCAST_BF16    1 = Trainer runs forward+loss under bf16 autocast
MI355X_NHWC             1 = Trainer converts model+batches to
                        channels_last (with AUTOCAST_BF16: the fastest
                        measured ResNet-50 config, profiles r02d)
MI355X_FORCE_DEV0       test-only: run a multi-rank world on ONE device
                        (IPC time-sharing) — the 8-GPU pre-flight
                        rehearsals; never set in production
MI355X_WORLD            world size for the spawn entrypoints off-GPU and
                        under MI355X_FORCE_DEV0
MI355X_CORE_DEBUG       reducer-core bisection aid (noop|norebind) — not
                        a production knob

RCCL's own tuning envs (NCCL_ALGO, NCCL_PROTO, NCCL_MIN/MAX_NCHANNELS)
pass straight through to the large-bucket collective path — the knobs
SURVEY §5.8 names for per-size algorithm selection over xGMI
(tools/bucket_sweep.py drives the sweep).
"""

import os


def env_flag(name: str, default: bool = False) -> bool:
    v = os.environ.get(name)
    if v is None:
        return default
    return v not in ("0", "false", "False", "")


def env_float(name: str, default: float) -> float:
    try:
        return float(os.environ.get(name, default))
    except ValueError:
        return default
