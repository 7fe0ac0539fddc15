"""Head-to-head vs stock PyTorch-ROCm on the SAME box.

Single-box A/B (numbers from different gpurun calls are never compared —
ROUND1_NOTES box-variance caveat): every arm runs in this one process.

    python tools/ab_stock.py --mode toy    [--steps 2000]
    python tools/ab_stock.py --mode resnet [--steps 20] [--dtype bf16]
        [--nhwc]

Arms:
  toy:    stock-eager (nn.Linear + F.mse_loss + torch.optim.SGD),
          stock-ddp-w1 (torch DDP over gloo, world 1),
          ours-hooks (generic autograd path: C++ autograd Functions +
          ReducerCore hooks + FusedSGD), ours-graph (whole-step hipGraph),
          ours-persistent (the flagship engine)
  resnet: stock-eager vs ours-hooks (multi-bucket reducer + fused SGD +
          hand CE), fp32 / bf16-autocast, NCHW / NHWC
Prints one JSON line per arm and persists gpurun_out/ab_<mode>.json.
"""
import argparse
import json
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from mi355x_ddp import ops  # noqa: E402
from mi355x_ddp.models import resnet50, toy_model  # noqa: E402
from mi355x_ddp.parallel import DDP, FusedSGD  # noqa: E402


def timeit(step, steps, warmup):
    for _ in range(warmup):
        step()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(steps):
        step()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / steps


def toy_arms(args, results):
    dev = torch.device("cuda", 0)
    x = torch.rand(32, 20, device=dev)
    t = torch.rand(32, 1, device=dev)

    # stock eager
    m = torch.nn.Linear(20, 1).to(dev)
    opt = torch.optim.SGD(m.parameters(), lr=1e-3)

    def stock():
        opt.zero_grad(set_to_none=False)
        torch.nn.functional.mse_loss(m(x), t).backward()
        opt.step()
    results["stock-eager"] = timeit(stock, args.steps, args.warmup)

    # stock DDP (world 1, gloo — the only stock-DDP form a 1-GPU box runs)
    try:
        sm = torch.nn.Linear(20, 1).to(dev)
        sddp = torch.nn.parallel.DistributedDataParallel(sm)
        sopt = torch.optim.SGD(sm.parameters(), lr=1e-3)

        def stock_ddp():
            sopt.zero_grad(set_to_none=False)
            torch.nn.functional.mse_loss(sddp(x), t).backward()
            sopt.step()
        results["stock-ddp-w1-gloo"] = timeit(stock_ddp, args.steps,
                                              args.warmup)
    except Exception as e:
        results["stock-ddp-w1-gloo"] = f"failed: {e!r}"

    # ours: generic hooks path (C++ core + C++ autograd Functions)
    om = toy_model(20, 1).to(dev)
    oeng = DDP(om, comm=None)
    oopt = FusedSGD(om.parameters(), lr=1e-3)
    oopt.attach_reducer(oeng.reducer)

    def ours():
        ops.mse_loss(oeng(x), t).backward()
        oeng.finalize_backward()
        oopt.step()
    results["ours-hooks"] = timeit(ours, args.steps, args.warmup)

    # ours: whole-step graph
    from mi355x_ddp.engine import GraphedAutogradStep, PersistentToyStep
    gm = toy_model(20, 1).to(dev)
    geng = DDP(gm, comm=None, cpp_hooks=False)
    gopt = FusedSGD(gm.parameters(), lr=1e-3)
    gopt.attach_reducer(geng.reducer)
    gs = GraphedAutogradStep(geng, ops.mse_loss, gopt,
                             finalize=geng.finalize_backward)
    results["ours-graph-1step"] = timeit(lambda: gs.step(x, t),
                                         args.steps, args.warmup)

    # ours: flagship persistent engine (single-step probes; the deferred
    # steady state is the bench headline, not reproduced here)
    pm = toy_model(20, 1).to(dev)
    pe = PersistentToyStep(pm, comm=None, lr=1e-3, use_mse=True)

    def pers():
        pe.step(x, t)
        pe.flush()
    results["ours-persistent-probe"] = timeit(pers, args.steps, args.warmup)


def resnet_arms(args, results):
    dev = torch.device("cuda", 0)
    bf16 = args.dtype == "bf16"
    fmt = torch.channels_last if args.nhwc else torch.contiguous_format
    x = torch.rand(args.batch, 3, 224, 224, device=dev) \
        .contiguous(memory_format=fmt)
    t = torch.rand(args.batch, 1000, device=dev)

    def autocast(cache=True):
        # cache_enabled=False is REQUIRED under graph capture (torch's
        # AMP+graphs rule: the autocast cast-cache frees its tensors
        # between iterations, which invalidates a capture)
        return torch.autocast("cuda", dtype=torch.bfloat16, enabled=bf16,
                              cache_enabled=cache)

    torch.manual_seed(0)
    m = resnet50().to(dev).to(memory_format=fmt)
    opt = torch.optim.SGD(m.parameters(), lr=1e-4)

    def stock():
        opt.zero_grad(set_to_none=False)
        with autocast():
            loss = torch.nn.functional.cross_entropy(m(x), t)
        loss.backward()
        opt.step()
    results["stock-eager"] = timeit(stock, args.steps, args.warmup)

    torch.manual_seed(0)
    om = resnet50().to(dev).to(memory_format=fmt)
    oeng = DDP(om, comm=None)
    oopt = FusedSGD(om.parameters(), lr=1e-4)
    oopt.attach_reducer(oeng.reducer)

    def ours():
        with autocast():
            loss = ops.cross_entropy(oeng(x), t)
        loss.backward()
        oeng.finalize_backward()
        oopt.step()
    results["ours-hooks"] = timeit(ours, args.steps, args.warmup)

    # whole-step hipGraph capture of the ResNet step (eager fallback if
    # MIOpen ops refuse capture — the arm then reports ~hooks numbers)
    from mi355x_ddp.engine import GraphedAutogradStep

    class _Autocast(torch.nn.Module):
        def __init__(self, inner):
            super().__init__()
            self.inner = inner

        def forward(self, xx):
            with autocast(cache=False):
                return self.inner(xx)

    torch.manual_seed(0)
    gm = resnet50().to(dev).to(memory_format=fmt)
    geng = DDP(gm, comm=None, cpp_hooks=False)
    gopt = FusedSGD(gm.parameters(), lr=1e-4)
    gopt.attach_reducer(geng.reducer)

    def gloss(y, tt):
        with autocast(cache=False):
            return ops.cross_entropy(y, tt)

    gs = GraphedAutogradStep(_Autocast(geng), gloss, gopt,
                             finalize=geng.finalize_backward,
                             warmup_steps=2)
    # trigger warmup+capture outside the timed region
    gs.step(x, t)
    results["ours-graph"] = timeit(lambda: gs.step(x, t),
                                   args.steps, args.warmup)
    results["graph_captured"] = bool(gs._graphs) and not gs._broken


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--mode", choices=["toy", "resnet"], required=True)
    ap.add_argument("--steps", type=int, default=None)
    ap.add_argument("--warmup", type=int, default=None)
    ap.add_argument("--batch", type=int, default=32)
    ap.add_argument("--dtype", choices=["fp32", "bf16"], default="fp32")
    ap.add_argument("--nhwc", action="store_true")
    args = ap.parse_args()
    if args.steps is None:
        args.steps = 2000 if args.mode == "toy" else 20
    if args.warmup is None:
        args.warmup = max(5, args.steps // 10)

    results = {}
    if args.mode == "toy":
        import torch.distributed as dist
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29793")
        dist.init_process_group("gloo", rank=0, world_size=1)
        toy_arms(args, results)
        dist.destroy_process_group()
    else:
        resnet_arms(args, results)

    out = {"mode": args.mode, "dtype": args.dtype, "nhwc": args.nhwc,
           "batch": args.batch, "steps": args.steps}
    out["us_per_step"] = {k: (v * 1e6 if isinstance(v, float) else v)
                          for k, v in results.items()}
    print(json.dumps(out), flush=True)
    os.makedirs("gpurun_out", exist_ok=True)
    tag = f"{args.mode}_{args.dtype}{'_nhwc' if args.nhwc else ''}"
    with open(f"gpurun_out/ab_{tag}.json", "w") as f:
        json.dump(out, f, indent=1)


if __name__ == "__main__":
    main()
