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
                        (hooks default | auto | fused | persistent |
                        graph; silently falls back to hooks when the
                        stage does not qualify)
MI355X_DTYPE            bf16 = cast the model in multigpu.py (BASELINE
                        config 2); unset = the reference's fp32

RCCL's own tuning envs (NCCL_ALGO, NCCL_PROTO, NCCL_MIN/MAX_NCHANNELS)
pass straight through to the large-bucket collective path — the knobs
SURVEY §5.8 names for per-size algorithm selection over xGMI.
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
