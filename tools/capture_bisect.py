"""Bisect whole-step hipGraph capture: which ingredient segfaults?
Run each variant in its own process: python tools/capture_bisect.py VARIANT
Variants: plain | ourops | hooks_py | hooks_cpp | nobwd"""
import sys
import torch

variant = sys.argv[1]
dev = "cuda:0"
torch.cuda.set_device(0)

if variant in ("plain", "nobwd"):
    m = torch.nn.Linear(64, 10).to(dev)
    loss_fn = torch.nn.functional.mse_loss
    opt = torch.optim.SGD(m.parameters(), lr=0.01)
    fin = lambda: None
elif variant == "ourops":
    sys.path.insert(0, ".")
    from mi355x_ddp import ops
    from mi355x_ddp.models import toy_model
    m = toy_model(64, 1).to(dev)
    loss_fn = ops.mse_loss
    opt = torch.optim.SGD(m.parameters(), lr=0.01)
    fin = lambda: None
else:  # hooks_py / hooks_cpp
    import os
    os.environ["MI355X_CPP_HOOKS"] = "1" if variant == "hooks_cpp" else "0"
    sys.path.insert(0, ".")
    from mi355x_ddp import ops
    from mi355x_ddp.parallel import DDP, FusedSGD
    m = torch.nn.Linear(64, 10).to(dev)
    eng = DDP(m, comm=None)
    assert (eng.reducer._core is not None) == (variant == "hooks_cpp")
    loss_fn = torch.nn.functional.mse_loss
    opt = FusedSGD(m.parameters(), lr=0.01)
    opt.attach_reducer(eng.reducer)
    fin = eng.finalize_backward
    m = eng

x = torch.randn(16, 64, device=dev)
t = torch.randn(16, 10 if variant != "ourops" else 1, device=dev)

def step():
    opt.zero_grad(set_to_none=False)
    loss = loss_fn(m(x), t)
    if variant != "nobwd":
        loss.backward()
        fin()
        opt.step()
    return loss

s = torch.cuda.Stream()
s.wait_stream(torch.cuda.current_stream())
with torch.cuda.stream(s):
    for _ in range(3):
        step()
torch.cuda.current_stream().wait_stream(s)
torch.cuda.synchronize()
print(f"[{variant}] warmup ok", flush=True)
g = torch.cuda.CUDAGraph()
with torch.cuda.graph(g):
    step()
print(f"[{variant}] capture ok", flush=True)
for _ in range(3):
    g.replay()
torch.cuda.synchronize()
print(f"[{variant}] replay ok", flush=True)
