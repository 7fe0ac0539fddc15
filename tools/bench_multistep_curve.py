"""Per-step cost of the multi-step trainer vs steps-per-launch S:
quantifies how launch/dispatch overhead amortizes (docs/KERNELS.md)."""
import sys
import time

import torch

sys.path.insert(0, __file__.rsplit("/", 2)[0])
from mi355x_ddp import ops
from mi355x_ddp.models import toy_model
from mi355x_ddp.parallel.reducer import Reducer

dev = "cuda:0"
torch.manual_seed(0)
m = toy_model(20, 1).to(dev)
red = Reducer(list(m.parameters()), comm=None)
b = red.buckets[0]
w_off = b.offsets[red._param_index[m.weight][1]]
b_off = b.offsets[red._param_index[m.bias][1]]
X = torch.rand(64 * 32, 20, device=dev)
T = torch.rand(64 * 32, 1, device=dev)
dummy = torch.Tensor()

print("| S (steps/launch) | fp32 ns/step | bf16 ns/step |")
print("|---|---|---|")
for S in (1, 2, 4, 8, 16, 32, 64):
    xs, ts = X[: S * 32], T[: S * 32]
    xb, tb = xs.bfloat16(), ts.bfloat16()
    pb = b.flat_param.bfloat16()
    reps = max(2000 // S, 100)
    row = [S]
    for x, t, p in ((xs, ts, b.flat_param), (xb, tb, pb)):
        for _ in range(20):
            ops.ext().toy_multistep(x, t, p, dummy, True, w_off, b_off,
                                    1e-4, 32)
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(reps):
            ops.ext().toy_multistep(x, t, p, dummy, True, w_off, b_off,
                                    1e-4, 32)
        torch.cuda.synchronize()
        row.append((time.perf_counter() - t0) / reps / S * 1e9)
    print(f"| {row[0]} | {row[1]:8.0f} | {row[2]:8.0f} |")
