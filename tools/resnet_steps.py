"""Profiling helper: N ResNet-50 training steps through the native engine
(multi-bucket reducer + fused SGD; conv/BN via MIOpen). Used under
rocprofv3 for the profile-stage kernel stats (SURVEY stage 5), and
standalone to time the step (MI355X_CHANNELS_LAST=1 tries NHWC, the
layout MIOpen's igemm kernels prefer on CDNA)."""
import os
import sys
import time

import torch

sys.path.insert(0, __file__.rsplit("/", 2)[0])
from mi355x_ddp import ops
from mi355x_ddp.models import resnet50
from mi355x_ddp.parallel import FusedSGD
from mi355x_ddp.parallel.reducer import Reducer

steps = int(sys.argv[1]) if len(sys.argv) > 1 else 20
nhwc = os.environ.get("MI355X_CHANNELS_LAST", "0") == "1"
torch.manual_seed(0)
dev = "cuda:0"
model = resnet50().to(dev)
x = torch.rand(32, 3, 224, 224, device=dev)
if nhwc:
    model = model.to(memory_format=torch.channels_last)
    x = x.contiguous(memory_format=torch.channels_last)
t = torch.rand(32, 1000, device=dev)
params = list(model.parameters())
reducer = Reducer(params, comm=None, bucket_cap_mb=25.0)
opt = FusedSGD(params, lr=1e-3)
opt.attach_reducer(reducer)

def one(n):
    for _ in range(n):
        loss = ops.cross_entropy(model(x), t)
        loss.backward()
        reducer.finalize()
        opt.step()
    torch.cuda.synchronize()
    return loss

one(5)  # warmup + MIOpen find
t0 = time.perf_counter()
loss = one(steps)
dt = (time.perf_counter() - t0) / steps
print(f"resnet50 bs32 {'nhwc' if nhwc else 'nchw'}: {dt*1e3:.2f} ms/step "
      f"({32/dt:.0f} img/s) loss={float(loss.detach()):.4f}")
