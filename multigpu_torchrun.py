"""Stage 3 — torchrun + fault-tolerant snapshot resume
(reference multigpu_torchrun.py).

Launch: `torchrun --standalone --nproc_per_node=N multigpu_torchrun.py
<total_epochs> <save_every> [snapshot_path]`.

Honors the torchrun env contract (RANK/LOCAL_RANK/WORLD_SIZE/MASTER_*,
SURVEY §2.2 N14) so the stock elastic agent supervises this engine: on a
worker failure torchrun restarts everyone, each worker finds snapshot.pt
and resumes from EPOCHS_RUN (reference flow).

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
    ddp_setup()  # env-var init (reference multigpu_torchrun.py:12-13)
    try:
        dataset, model, optimizer = load_train_objs()
        train_data = prepare_dataloader(dataset, batch_size=32, distributed=True)
        device = None if torch.cuda.is_available() else "cpu"  # None -> LOCAL_RANK
        trainer = Trainer(model, train_data, optimizer, device, save_every,
                          snapshot_path=snapshot_path,
                          engine=os.environ.get("MI355X_ENGINE", "hooks"))
        trainer.train(total_epochs)
    finally:
        destroy_process_group()


if __name__ == "__main__":
    total_epochs = int(sys.argv[1])
    save_every = int(sys.argv[2])
    snapshot = sys.argv[3] if len(sys.argv) > 3 else "snapshot.pt"
    main(total_epochs, save_every, snapshot)
