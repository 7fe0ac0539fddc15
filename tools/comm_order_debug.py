"""Does creating the RCCL comm BEFORE the first MIOpen find change the
chosen conv solutions? arg: commfirst | modelfirst"""
import os, sys, time
import torch
import torch.distributed as dist
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from mi355x_ddp import ops
from mi355x_ddp.models import resnet50
from mi355x_ddp.parallel import FusedSGD
from mi355x_ddp.parallel.comm import RcclCommAdapter
from mi355x_ddp.parallel.reducer import Reducer

order = sys.argv[1]
os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
os.environ.setdefault("MASTER_PORT", "29795")
dist.init_process_group("gloo", rank=0, world_size=1)
torch.cuda.set_device(0)
dev = torch.device("cuda", 0)
comm = RcclCommAdapter(dev) if order == "commfirst" else None
x = torch.rand(32, 3, 224, 224, device=dev)
t = torch.rand(32, 1000, device=dev)
torch.manual_seed(0)
m = resnet50().to(dev)
params = list(m.parameters())
red = Reducer(params, comm=comm, bucket_cap_mb=25.0)
opt = FusedSGD(params, lr=1e-4)
opt.attach_reducer(red)
def one(n):
    for _ in range(n):
        ops.cross_entropy(m(x), t).backward()
        red.finalize()
        opt.step()
    torch.cuda.synchronize()
one(8)
t0 = time.perf_counter(); one(15)
print(f"{order}: {(time.perf_counter()-t0)/15*1e3:.2f} ms/step", flush=True)
dist.destroy_process_group()
