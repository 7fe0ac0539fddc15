"""Round 2 of the capture bisect: which part of the C++ hook segfaults
hipGraph capture_end? Usage: python tools/capture_bisect2.py VARIANT
Variants: noop | norebind | full | relaxed | thread_local"""
import os
import sys

variant = sys.argv[1]
if variant in ("noop", "norebind"):
    os.environ["MI355X_CORE_DEBUG"] = variant
import torch  # noqa: E402

sys.path.insert(0, ".")
from mi355x_ddp.parallel import DDP, FusedSGD  # noqa: E402

dev = "cuda:0"
torch.cuda.set_device(0)
m = torch.nn.Linear(64, 10).to(dev)
eng = DDP(m, comm=None)
assert eng.reducer._core is not None
opt = FusedSGD(m.parameters(), lr=0.01)
opt.attach_reducer(eng.reducer)
x = torch.randn(16, 64, device=dev)
t = torch.randn(16, 10, device=dev)

def step():
    loss = torch.nn.functional.mse_loss(eng(x), t)
    loss.backward()
    eng.finalize_backward()
    opt.step()

s = torch.cuda.Stream()
s.wait_stream(torch.cuda.current_stream())
with torch.cuda.stream(s):
    for _ in range(3):
        step()
torch.cuda.current_stream().wait_stream(s)
torch.cuda.synchronize()
print(f"[{variant}] warmup ok", flush=True)
g = torch.cuda.CUDAGraph()
mode = {"relaxed": "relaxed", "thread_local": "thread_local"}.get(variant, "global")
with torch.cuda.graph(g, capture_error_mode=mode):
    step()
print(f"[{variant}] capture ok", flush=True)
for _ in range(3):
    g.replay()
torch.cuda.synchronize()
print(f"[{variant}] replay ok", flush=True)
