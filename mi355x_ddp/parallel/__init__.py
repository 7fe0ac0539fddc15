"""Native data-parallel engine: communicator, reducer, DDP wrapper, fused
optimizer (§2.3 — DP is the reference's one
parallelism strategy, re-implemented here from scratch for MI355X)."""

from .comm import (ddp_setup, create_comm, GlooComm,  # noqa: F401
                   P2pMeshComm, RcclCommAdapter)
from .reducer import Reducer, Bucket  # noqa: F401
from .ddp import DDP  # noqa: F401
from .sgd import FusedSGD  # noqa: F401
