"""Stage 4 — multi-node torchrun via SLURM (reference multinode_torchrun.py).

Launch per node (see slurm/sbatch_run.sh):
`torchrun --nnodes N --nproc_per_node G --rdzv_backend c10d
 --rdzv_endpoint head:29500 multinode_torchrun.py <total_epochs> <save_every>`

Differences from stage 3, kept for parity: MSE loss
(reference multinode_torchrun.py:46) and the banner shows the GLOBAL rank
(reference :25,52). The reference's snapshot race (every node's local rank
0 writing the shared snapshot, :68) is fixed: only global rank 0 saves.

MI355X_ENGINE=persistent|fused|auto engages the fast toy engines through this
entrypoint (silently falls back to the generic hooks path when the
model/loss/device do not qualify — e.g. on CPU, or for the CE-loss stages).
"""

import os
import sys

import torch
from torch.distributed import destroy_process_group

from mi355x_ddp.data import ToyDataset, prepare_dataloader
from mi355x_ddp.models import toy_model
from mi355x_ddp.parallel import FusedSGD, ddp_setup
from mi355x_ddp.trainer import Trainer


def load_train_objs():
    train_set = ToyDataset(2048)
    model = toy_model(20, 1)
    optimizer = FusedSGD(model.parameters(), lr=1e-3)
    return train_set, model, optimizer


def main(total_epochs: int, save_every: int, snapshot_path: str = "snapshot.pt"):
    ddp_setup()
    try:
        dataset, model, optimizer = load_train_objs()
        train_data = prepare_dataloader(dataset, batch_size=32, distributed=True)
        device = None if torch.cuda.is_available() else "cpu"
        trainer = Trainer(model, train_data, optimizer, device, save_every,
                          snapshot_path=snapshot_path, loss_fn="mse",
                          engine=os.environ.get("MI355X_ENGINE", "hooks"))
        trainer.train(total_epochs)
    finally:
        destroy_process_group()


if __name__ == "__main__":
    total_epochs = int(sys.argv[1])
    save_every = int(sys.argv[2])
    snapshot = sys.argv[3] if len(sys.argv) > 3 else "snapshot.pt"
    main(total_epochs, save_every, snapshot)
