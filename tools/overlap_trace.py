"""Comm/compute overlap evidence (VERDICT r01 item 4 / missing #2).

Runs ResNet-50 training steps through the multi-bucket reducer with a
REAL RCCL communicator, records a Kineto (roctracer) trace, and measures
— from the trace itself — how much RCCL all-reduce kernel time runs
CONCURRENTLY with backward compute kernels on the other stream. Prints a
JSON summary and keeps the chrome trace for inspection:

    python tools/overlap_trace.py [--steps 8] [--out-dir gpurun_out/overlap]

Works at world 1 (self-RCCL: the collective still runs as an RCCL device
kernel on the dedicated comm stream, ordered by the same events as the
multi-GPU path — exactly the machinery whose overlap is in question) and
under torchrun at any world size on a multi-GPU node.
"""
import argparse
import json
import os
import sys

import torch
import torch.distributed as dist

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from mi355x_ddp import ops  # noqa: E402
from mi355x_ddp.models import resnet50  # noqa: E402
from mi355x_ddp.parallel import FusedSGD  # noqa: E402
from mi355x_ddp.parallel.comm import RcclCommAdapter  # noqa: E402
from mi355x_ddp.parallel.reducer import Reducer  # noqa: E402


def overlap_from_trace(path):
    """Parse the chrome trace: total RCCL kernel time and the fraction of
    it that overlaps (wall-clock) a non-RCCL GPU kernel."""
    with open(path) as f:
        tr = json.load(f)
    evs = [e for e in tr.get("traceEvents", [])
           if e.get("ph") == "X" and e.get("cat") in ("kernel", "gpu_op",
                                                      "Kernel")]
    rccl, comp = [], []
    for e in evs:
        n = e.get("name", "").lower()
        # RCCL device kernels: multi-rank "ncclDevKernel_*"; world-1
        # reductions dispatch as "oneRankReduce<...>" (measured on-box)
        is_rccl = "nccl" in n or "rccl" in n or "onerankreduce" in n
        (rccl if is_rccl else comp).append((e["ts"], e["ts"] + e["dur"]))
    comp.sort()
    # merge compute intervals
    merged = []
    for s, e in comp:
        if merged and s <= merged[-1][1]:
            merged[-1][1] = max(merged[-1][1], e)
        else:
            merged.append([s, e])

    def olap(s, e):
        tot = 0.0
        for ms, me in merged:
            lo, hi = max(s, ms), min(e, me)
            if lo < hi:
                tot += hi - lo
        return tot

    rccl_total = sum(e - s for s, e in rccl)
    rccl_olap = sum(olap(s, e) for s, e in rccl)
    comp_total = sum(e - s for s, e in merged)
    return {"rccl_kernels": len(rccl),
            "rccl_total_us": rccl_total,
            "rccl_overlapped_us": rccl_olap,
            "overlap_fraction": rccl_olap / rccl_total if rccl_total else 0.0,
            "compute_total_us": comp_total}


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--steps", type=int, default=8)
    ap.add_argument("--batch", type=int, default=32)
    ap.add_argument("--cap-mb", type=float, default=25.0)
    ap.add_argument("--out-dir", default="gpurun_out/overlap")
    args = ap.parse_args()

    rank = int(os.environ.get("RANK", 0))
    world = int(os.environ.get("WORLD_SIZE", 1))
    local = int(os.environ.get("LOCAL_RANK", 0))
    if world > 1:
        dist.init_process_group("cpu:gloo,cuda:nccl")
    else:
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29792")
        dist.init_process_group("gloo", rank=0, world_size=1)
    torch.cuda.set_device(local)
    device = torch.device("cuda", local)

    torch.manual_seed(0)
    model = resnet50().to(device)
    # MIOpen find BEFORE RCCL init: an initialized RCCL communicator
    # during the first conv finds makes MIOpen pick ~2.5x slower conv
    # solutions (measured: 44.3 vs 17.5 ms/step, tools/comm_order_debug).
    # Warm every fwd+bwd conv shape first, then bring up the comm.
    xw = torch.rand(args.batch, 3, 224, 224, device=device)
    tw = torch.rand(args.batch, 1000, device=device)
    for _ in range(2):
        ops.cross_entropy(model(xw), tw).backward()
    for p in model.parameters():
        p.grad = None
    torch.cuda.synchronize()
    comm = RcclCommAdapter(device)
    params = list(model.parameters())
    red = Reducer(params, comm=comm, bucket_cap_mb=args.cap_mb)
    red.broadcast_params(0)
    opt = FusedSGD(params, lr=1e-4)
    opt.attach_reducer(red)
    x = torch.rand(args.batch, 3, 224, 224, device=device)
    t = torch.rand(args.batch, 1000, device=device)

    def one(n):
        for _ in range(n):
            ops.cross_entropy(model(x), t).backward()
            red.finalize()
            opt.step()
        torch.cuda.synchronize()

    one(4)  # warmup
    os.makedirs(args.out_dir, exist_ok=True)
    from torch.profiler import ProfilerActivity, profile
    with profile(activities=[ProfilerActivity.CPU, ProfilerActivity.CUDA]) \
            as prof:
        one(args.steps)
    trace = os.path.join(args.out_dir, f"overlap_rank{rank}.json")
    prof.export_chrome_trace(trace)
    summary = overlap_from_trace(trace)
    summary.update(workload="resnet50 fp32", steps=args.steps,
                   world=world, cap_mb=args.cap_mb,
                   buckets=len(red.buckets))
    if rank == 0:
        print(json.dumps(summary), flush=True)
        with open(os.path.join(args.out_dir, "summary.json"), "w") as f:
            json.dump(summary, f, indent=1)
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
