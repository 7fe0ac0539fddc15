"""Mechanism probe for the RCCL-before-MIOpen-find regression.
arg: stockpg | noprio | stream | initdestroy | findnormal"""
import os, sys, time
v = sys.argv[1]
if v == "noprio":
    os.environ["MI355X_COMM_PRIO"] = "0"
if v == "findnormal":
    os.environ["MIOPEN_FIND_MODE"] = "1"
import torch
import torch.distributed as dist
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from mi355x_ddp import ops
from mi355x_ddp.models import resnet50
from mi355x_ddp.parallel import FusedSGD
from mi355x_ddp.parallel.reducer import Reducer

os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
os.environ.setdefault("MASTER_PORT", "29796")
torch.cuda.set_device(0)
dev = torch.device("cuda", 0)
comm = None
if v == "stockpg":
    # c10d ProcessGroupNCCL world-1: does stock torch's own RCCL init
    # trigger the same regression?
    dist.init_process_group("nccl", rank=0, world_size=1)
    t0 = torch.ones(4, device=dev)
    dist.all_reduce(t0)
    torch.cuda.synchronize()
else:
    dist.init_process_group("gloo", rank=0, world_size=1)
    if v in ("noprio", "initdestroy", "findnormal"):
        from mi355x_ddp.parallel.comm import RcclCommAdapter
        comm = RcclCommAdapter(dev)
        if v == "initdestroy":
            del comm._comm
            comm = None
    elif v == "stream":
        _s = torch.cuda.Stream(priority=-1)  # just a high-prio stream

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
print(f"{v}: {(time.perf_counter()-t0)/15*1e3:.2f} ms/step", flush=True)
dist.destroy_process_group()
