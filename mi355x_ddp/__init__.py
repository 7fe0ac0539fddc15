"""mi355x_ddp — an MI355X-native distributed data-parallel training harness.

Re-creates the capability surface of the PyTorch DDP tutorial series
(`subramen/distributed-pytorch`, mounted read-only at /root/reference): the
`Trainer` API (reference single_gpu.py:6-45 and friends), the five-stage
entrypoint progression, the torchrun env contract and the `snapshot.pt`
checkpoint format — with the native machinery the tutorial borrows from
PyTorch's internals (DDP reducer, collectives, hot kernels, fused optimizer)
re-built from scratch for CDNA4/gfx950:

- hand-written HIP kernels (MFMA-tiled linear fwd/bwd, CE/MSE loss, gradient
  flatten/unflatten, fused SGD) in `mi355x_ddp.ops`
- an RCCL-over-xGMI communicator driven directly (not via ProcessGroupNCCL)
  in `mi355x_ddp.parallel.comm`
- a bucketed, backward-overlapped gradient reducer in
  `mi355x_ddp.parallel.reducer` / `.ddp`
"""

__version__ = "0.1.0"

from . import data  # noqa: F401
from .trainer import Trainer  # noqa: F401
