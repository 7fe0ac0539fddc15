"""Time ViT-L/32 training steps through the native engine (multi-bucket
reducer + fused SGD; the model the reference imports but leaves commented
out — multigpu_profile.py:24). MI355X_AUTOCAST_BF16=1 for bf16."""
import os
import sys
import time

import torch

sys.path.insert(0, __file__.rsplit("/", 2)[0])
from mi355x_ddp import ops
from mi355x_ddp.models import vit_l_32
from mi355x_ddp.parallel import FusedSGD
from mi355x_ddp.parallel.reducer import Reducer

steps = int(sys.argv[1]) if len(sys.argv) > 1 else 20
bf16 = os.environ.get("MI355X_AUTOCAST_BF16") == "1"
torch.manual_seed(0)
dev = "cuda:0"
model = vit_l_32().to(dev)
x = torch.rand(32, 3, 224, 224, device=dev)
t = torch.rand(32, 1000, device=dev)
params = list(model.parameters())
reducer = Reducer(params, comm=None, bucket_cap_mb=25.0)
opt = FusedSGD(params, lr=1e-4)
opt.attach_reducer(reducer)


def one(n):
    for _ in range(n):
        if bf16:
            with torch.autocast("cuda", dtype=torch.bfloat16):
                loss = ops.cross_entropy(model(x), t)
        else:
            loss = ops.cross_entropy(model(x), t)
        loss.backward()
        reducer.finalize()
        opt.step()
    torch.cuda.synchronize()
    return loss


one(5)
t0 = time.perf_counter()
loss = one(steps)
dt = (time.perf_counter() - t0) / steps
print(f"vit_l_32 bs32 {'bf16' if bf16 else 'fp32'}: {dt*1e3:.2f} ms/step "
      f"({32/dt:.0f} img/s) buckets={len(reducer.buckets)} "
      f"loss={float(loss.detach()):.4f}")
