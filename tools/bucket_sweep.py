"""Bucket-cap autotune harness.

Sweeps MI355X_BUCKET_MB over the multi-bucket (ResNet-50) reducer path
with REAL RCCL collectives on the comm stream and persists step times:

    python tools/bucket_sweep.py [--caps 4,8,16,25,32,64,128] [--steps 30]
        [--out gpurun_out/bucket_sweep.json]

World handling: run standalone on one GPU (a world-1 RCCL communicator —
the all-reduce is a device-local pass through the full RCCL machinery:
comm-stream launch, event fencing, per-bucket kernels), or under torchrun
at any world size on a multi-GPU node, where the same script measures the
real xGMI exchange:

    torchrun --standalone --nproc_per_node=8 tools/bucket_sweep.py

RCCL algorithm/protocol presets (NCCL_ALGO / NCCL_PROTO) are inherited
from the environment so the same harness drives an algo sweep on real
hardware:  for a in Ring Tree; do NCCL_ALGO=$a torchrun ... ; done
(per-size algorithm choice is a >1-GPU property; nothing to measure at
world 1 — see profiles/README.md).
"""
import argparse
import json
import os
import sys
import time

import torch
import torch.distributed as dist

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from mi355x_ddp import ops  # noqa: E402
from mi355x_ddp.models import resnet50  # noqa: E402
from mi355x_ddp.parallel import FusedSGD  # noqa: E402
from mi355x_ddp.parallel.comm import RcclCommAdapter  # noqa: E402
from mi355x_ddp.parallel.reducer import Reducer  # noqa: E402


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--caps", default="4,8,16,25,32,64,128")
    ap.add_argument("--steps", type=int, default=30)
    ap.add_argument("--warmup", type=int, default=8)
    ap.add_argument("--batch", type=int, default=32)
    ap.add_argument("--out", default="gpurun_out/bucket_sweep.json")
    args = ap.parse_args()

    rank = int(os.environ.get("RANK", 0))
    world = int(os.environ.get("WORLD_SIZE", 1))
    local = int(os.environ.get("LOCAL_RANK", 0))
    if world > 1:
        dist.init_process_group("cpu:gloo,cuda:nccl")
    else:
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29791")
        dist.init_process_group("gloo", rank=0, world_size=1)
    torch.cuda.set_device(local)
    device = torch.device("cuda", local)

    torch.manual_seed(0)
    x = torch.rand(args.batch, 3, 224, 224, device=device)
    t = torch.rand(args.batch, 1000, device=device)

    # MIOpen find BEFORE RCCL init (see tools/overlap_trace.py / the
    # comm_order_debug measurement: find with a live communicator picks
    # ~2.5x slower conv solutions)
    warm = resnet50().to(device)
    for _ in range(2):
        ops.cross_entropy(warm(x), t).backward()
    del warm
    torch.cuda.empty_cache()
    torch.cuda.synchronize()
    comm = RcclCommAdapter(device)

    results = []
    for cap in [float(c) for c in args.caps.split(",")]:
        torch.manual_seed(0)
        model = resnet50().to(device)
        params = list(model.parameters())
        red = Reducer(params, comm=comm, bucket_cap_mb=cap)
        red.broadcast_params(0)
        opt = FusedSGD(params, lr=1e-4)
        opt.attach_reducer(red)

        def one(n):
            for _ in range(n):
                ops.cross_entropy(model(x), t).backward()
                red.finalize()
                opt.step()
            torch.cuda.synchronize()

        one(args.warmup)
        comm.barrier()
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        one(args.steps)
        comm.barrier()
        dt = (time.perf_counter() - t0) / args.steps
        row = {"cap_mb": cap, "buckets": len(red.buckets),
               "ms_per_step": dt * 1e3,
               "img_per_s_per_gpu": args.batch / dt,
               "world": world,
               "nccl_algo": os.environ.get("NCCL_ALGO", ""),
               "nccl_proto": os.environ.get("NCCL_PROTO", "")}
        results.append(row)
        if rank == 0:
            print(json.dumps(row), flush=True)
        red.detach_hooks()
        del model, params, red, opt
        torch.cuda.empty_cache()

    if rank == 0:
        os.makedirs(os.path.dirname(args.out) or ".", exist_ok=True)
        with open(args.out, "w") as f:
            json.dump({"workload": "resnet50 fp32 train step",
                       "batch": args.batch, "steps": args.steps,
                       "world": world, "results": results}, f, indent=1)
        best = min(results, key=lambda r: r["ms_per_step"])
        print(f"# best: cap={best['cap_mb']} MB "
              f"({best['ms_per_step']:.2f} ms/step)", flush=True)
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
