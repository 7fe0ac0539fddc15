"""Operator microbenchmarks: the hand-written kernels vs their
PyTorch-ROCm (rocBLAS/ATen) equivalents on the shapes the framework
actually runs. Emits a markdown table; run on a GPU box, output committed
under profiles/.

Usage: python tools/bench_kernels.py [iters]
"""
import sys
import time

import torch

sys.path.insert(0, __file__.rsplit("/", 2)[0])
from mi355x_ddp import ops

DEV = "cuda:0"
ITERS = int(sys.argv[1]) if len(sys.argv) > 1 else 2000


def timeit(fn, iters=ITERS):
    for _ in range(50):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6  # us


def row(name, ours_us, ref_us, note=""):
    print(f"| {name} | {ours_us:8.2f} | {ref_us:8.2f} | "
          f"{ref_us / ours_us:5.2f}x | {note} |")


def main():
    torch.manual_seed(0)
    print("| op (shape) | ours µs | torch µs | speedup | note |")
    print("|---|---|---|---|---|")

    # toy linear fwd (32x20 @ 20x1)
    x = torch.rand(32, 20, device=DEV)
    w = torch.rand(1, 20, device=DEV)
    b = torch.rand(1, device=DEV)
    row("linear fwd 32x20->1", timeit(lambda: ops.ext().linear_fwd(x, w, b)),
        timeit(lambda: torch.nn.functional.linear(x, w, b)), "toy shape")

    # FC fwd (ResNet head): 32x2048 @ 2048x1000
    xf = torch.rand(32, 2048, device=DEV)
    wf = torch.rand(1000, 2048, device=DEV)
    bf = torch.rand(1000, device=DEV)
    row("linear fwd 32x2048->1000 (raw split-K kernel)",
        timeit(lambda: ops.ext().linear_fwd(xf, wf, bf)),
        timeit(lambda: torch.nn.functional.linear(xf, wf, bf)),
        "ResNet FC, rocBLAS ref")
    row("linear fwd 32x2048->1000 (ops.linear, routed)",
        timeit(lambda: ops.linear(xf, wf, bf)),
        timeit(lambda: torch.nn.functional.linear(xf, wf, bf)),
        "library-shape -> rocBLAS")

    # FC dX: 32x1000 @ 1000x2048
    dy = torch.rand(32, 1000, device=DEV)
    row("linear bwd dX 32x1000x2048 (split-N)",
        timeit(lambda: ops.ext().linear_bwd_input(dy, wf)),
        timeit(lambda: dy @ wf), "")

    # FC dW: 1000x32 @ 32x2048
    dw = torch.empty(1000, 2048, device=DEV)
    db = torch.empty(1000, device=DEV)
    row("linear bwd dW+db 1000x2048",
        timeit(lambda: ops.ext().linear_bwd_weight(xf, dy, dw, db, False)),
        timeit(lambda: (dy.t() @ xf, dy.sum(0))), "fused db")

    # bf16 GEMM 128x256x256
    xb = torch.rand(128, 256, device=DEV).bfloat16()
    wb = torch.rand(256, 256, device=DEV).bfloat16()
    bb = torch.rand(256, device=DEV).bfloat16()
    row("bf16 GEMM 128x256x256 (16x16x32 MFMA)",
        timeit(lambda: ops.ext().gemm_bf16(xb, wb, bb)),
        timeit(lambda: torch.nn.functional.linear(xb, wb, bb)), "")

    # CE fwd+bwd (B=32, C=1000)
    y = torch.rand(32, 1000, device=DEV)
    t = torch.softmax(torch.rand(32, 1000, device=DEV), 1)

    def ce_ours():
        loss, probs, tsum = ops.ext().ce_fwd(y, t)
        ops.ext().ce_bwd(probs, t, tsum, 1.0)

    yl = y.clone().requires_grad_(True)

    def ce_torch():
        loss = torch.nn.functional.cross_entropy(yl, t)
        g, = torch.autograd.grad(loss, yl)

    row("cross-entropy fwd+bwd 32x1000", timeit(ce_ours), timeit(ce_torch), "")

    # fused SGD over 25 MB bucket
    p = torch.rand(6_250_000, device=DEV)
    g = torch.rand(6_250_000, device=DEV)
    row("SGD+zero_grad 25 MB bucket",
        timeit(lambda: ops.ext().sgd_flat(p, g, 1e-3, True)),
        timeit(lambda: (p.add_(g, alpha=-1e-3), g.zero_())),
        "1 launch vs 2")

    # epoch shard vs randperm+index
    X = torch.rand(2048, 20, device=DEV)
    T = torch.rand(2048, 1, device=DEV)

    def shard_torch():
        perm = torch.randperm(2048, device=DEV)
        return X[perm].contiguous(), T[perm].contiguous()

    row("epoch shard 2048x21 (in-kernel perm)",
        timeit(lambda: ops.ext().epoch_shard(X, T, 7, 0, 1), 500),
        timeit(shard_torch, 500), "vs randperm+2x index")

    # whole train step: persistent multistep (per step) vs fused vs torch
    from mi355x_ddp.engine import PersistentToyStep, ToyFusedStep
    from mi355x_ddp.models import toy_model
    m1 = toy_model(20, 1).to(DEV)
    eng = PersistentToyStep(m1, comm=None, lr=1e-3)
    Xs = torch.rand(64 * 32, 20, device=DEV)
    Ts = torch.rand(64 * 32, 1, device=DEV)
    eng.bind_shard(Xs, Ts, 32)

    def step64():
        for i in range(64):
            eng.step_shard(i)
        eng.flush()

    us64 = timeit(step64, 200) / 64

    m2 = toy_model(20, 1).to(DEV)
    fused = ToyFusedStep(m2, comm=None, lr=1e-3)
    xs0, ts0 = Xs[:32], Ts[:32]
    usf = timeit(lambda: fused.step(xs0, ts0))

    m3 = torch.nn.Linear(20, 1).to(DEV)
    opt = torch.optim.SGD(m3.parameters(), lr=1e-3)

    def torch_step():
        opt.zero_grad(set_to_none=False)
        torch.nn.functional.mse_loss(m3(xs0), ts0).backward()
        opt.step()

    ust = timeit(torch_step)
    row("toy TRAIN STEP (persistent, per step)", us64, ust, "vs torch eager")
    row("toy TRAIN STEP (fused single launch)", usf, ust, "vs torch eager")


if __name__ == "__main__":
    main()
