"""Debug: same process, resnet step time with comm=None vs world-1 RCCL."""
import os, sys, time
import torch
import torch.distributed as dist
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from mi355x_ddp import ops
from mi355x_ddp.models import resnet50
from mi355x_ddp.parallel import FusedSGD
from mi355x_ddp.parallel.comm import RcclCommAdapter
from mi355x_ddp.parallel.reducer import Reducer

os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
os.environ.setdefault("MASTER_PORT", "29794")
dist.init_process_group("gloo", rank=0, world_size=1)
torch.cuda.set_device(0)
dev = torch.device("cuda", 0)
x = torch.rand(32, 3, 224, 224, device=dev)
t = torch.rand(32, 1000, device=dev)

def run(comm, tag, hooks_env):
    os.environ["MI355X_CPP_HOOKS"] = hooks_env
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
    print(f"{tag}: {(time.perf_counter()-t0)/15*1e3:.2f} ms/step", flush=True)
    red.detach_hooks()
    del m, params, red, opt
    torch.cuda.empty_cache()

run(None, "no-comm cpp-hooks", "1")
run(None, "no-comm py-hooks ", "0")
comm = RcclCommAdapter(dev)
run(comm, "rccl-w1 cpp-hooks", "1")
run(comm, "rccl-w1 py-hooks ", "0")
run(None, "no-comm again    ", "1")
dist.destroy_process_group()
