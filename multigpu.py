"""Stage 2 — mp.spawn DDP on one node (reference multigpu.py).

Same CLI (`python multigpu.py <total_epochs> <save_every>`); spawns one
process per GPU; the DDP engine is the native reducer + RCCL-over-xGMI
(mi355x_ddp.parallel), not ProcessGroupNCCL. The reference's argument-order
bug (multigpu.py:82 vs :96 — save_every/total_epochs swapped at the spawn
site) is fixed.

Set MI355X_DTYPE=bf16 for the bf16 variant (BASELINE.json config 2).
CPU fallback (no GPU): world size from MI355X_WORLD (default 2) over gloo —
used by the CPU test tier.

MI355X_ENGINE=persistent|fused|auto engages the fast toy engines through this
entrypoint (silently falls back to the generic hooks path when the
model/loss/device do not qualify — e.g. on CPU, or for the CE-loss stages).
"""

import os
import sys

import torch
import torch.multiprocessing as mp
from torch.distributed import destroy_process_group

from mi355x_ddp.data import ToyDataset, prepare_dataloader
from mi355x_ddp.models import toy_model
from mi355x_ddp.parallel import FusedSGD, ddp_setup
from mi355x_ddp.trainer import Trainer


def load_train_objs():
    train_set = ToyDataset(2048)
    model = toy_model(20, 1)
    if os.environ.get("MI355X_DTYPE") == "bf16":
        model = model.to(torch.bfloat16)
    optimizer = FusedSGD(model.parameters(), lr=1e-3)
    return train_set, model, optimizer


def main(rank: int, world_size: int, total_epochs: int, save_every: int):
    ddp_setup(rank, world_size)
    try:
        dataset, model, optimizer = load_train_objs()
        train_data = prepare_dataloader(dataset, batch_size=32,
                                        distributed=True,
                                        num_replicas=world_size, rank=rank)
        device = rank if torch.cuda.is_available() else "cpu"
        trainer = Trainer(model, train_data, optimizer, device, save_every,
                          engine=os.environ.get("MI355X_ENGINE", "hooks"))
        trainer.train(total_epochs)
    finally:
        destroy_process_group()


if __name__ == "__main__":
    total_epochs = int(sys.argv[1])
    save_every = int(sys.argv[2])
    if os.environ.get("MI355X_FORCE_DEV0") == "1":
        # pre-flight rehearsal: N ranks time-sharing one device via IPC
        world_size = int(os.environ.get("MI355X_WORLD", 2))
    elif torch.cuda.is_available():
        world_size = torch.cuda.device_count()
    else:
        world_size = int(os.environ.get("MI355X_WORLD", 2))
    mp.spawn(main, args=(world_size, total_epochs, save_every),
             nprocs=world_size)
