"""After a failed (invalidated) capture + recovery variant, can the
process capture again? variant: both | noend | norelease | none"""
import os
import sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch

variant = sys.argv[1]
torch.cuda.set_device(0)
dev = "cuda:0"
m = torch.nn.Linear(32, 8).to(dev)
x = torch.randn(4, 32, device=dev)

def step():
    return m(x).sum()

# warm
s = torch.cuda.Stream(); s.wait_stream(torch.cuda.current_stream())
with torch.cuda.stream(s):
    for _ in range(3):
        step().backward()
torch.cuda.current_stream().wait_stream(s)
torch.cuda.synchronize()

g = torch.cuda.CUDAGraph()
ctx = torch.cuda.graph(g)
try:
    with ctx:
        step()
        torch.cuda.synchronize()  # poison: invalidates capture
except Exception as e:
    print("capture failed as planned:", type(e).__name__, flush=True)
    try:
        ctx.stream_ctx.__exit__(None, None, None)
    except Exception:
        pass
    d = torch.cuda.current_device()
    if variant in ("both", "norelease"):
        try: torch._C._cuda_endAllocateToPool(d, ctx.pool[0])
        except Exception as e2: print("end:", e2, flush=True)
    if variant in ("both", "noend"):
        try: torch._C._cuda_releasePool(d, ctx.pool[0])
        except Exception as e2: print("rel:", e2, flush=True)
    from mi355x_ddp import ops
    ops.ext().clear_hip_errors()
    dg = torch.cuda.default_generators[d]
    fresh = torch.Generator(device=f"cuda:{d}")
    fresh.set_state(dg.get_state())
    dg.graphsafe_set_state(fresh)
    torch.cuda.synchronize()

print("eager after recovery:", float(step()), flush=True)
print("randn ok:", torch.randn(4, device=dev).shape, flush=True)

g2 = torch.cuda.CUDAGraph()
with torch.cuda.graph(g2):
    y = step()
g2.replay()
torch.cuda.synchronize()
print(f"[{variant}] recapture ok", flush=True)
