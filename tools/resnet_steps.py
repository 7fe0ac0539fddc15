"""Profiling helper: N ResNet-50 training steps through the native engine
(multi-bucket reducer + fused SGD; conv/BN via MIOpen). Used under
rocprofv3 to collect the profile-stage kernel stats (SURVEY stage 5)."""
import sys

import torch

sys.path.insert(0, __file__.rsplit("/", 2)[0])
from mi355x_ddp import ops
from mi355x_ddp.models import resnet50
from mi355x_ddp.parallel import FusedSGD
from mi355x_ddp.parallel.reducer import Reducer

steps = int(sys.argv[1]) if len(sys.argv) > 1 else 20
torch.manual_seed(0)
dev = "cuda:0"
model = resnet50().to(dev)
params = list(model.parameters())
reducer = Reducer(params, comm=None, bucket_cap_mb=25.0)
opt = FusedSGD(params, lr=1e-3)
opt.attach_reducer(reducer)
x = torch.rand(32, 3, 224, 224, device=dev)
t = torch.rand(32, 1000, device=dev)
for s in range(steps):
    loss = ops.cross_entropy(model(x), t)
    loss.backward()
    reducer.finalize()
    opt.step()
torch.cuda.synchronize()
print("resnet steps done:", steps, "loss:", float(loss.detach()))
