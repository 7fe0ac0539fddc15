"""Stage 1 — single-process training (reference single_gpu.py).

Same CLI (`python single_gpu.py <total_epochs> <save_every>`), same banner,
same `checkpoint.pt` raw-state_dict format. Runs on GPU 0 when a GPU is
present, else on CPU (BASELINE.json config 1: the CPU plumbing path).

MI355X_ENGINE=persistent|fused|auto engages the fast toy engines through this
entrypoint (silently falls back to the generic hooks path when the
model/loss/device do not qualify — e.g. on CPU, or for the CE-loss stages).
"""

import os
import sys

import torch

from mi355x_ddp.data import ToyDataset, prepare_dataloader
from mi355x_ddp.models import toy_model
from mi355x_ddp.parallel import FusedSGD
from mi355x_ddp.trainer import Trainer


def load_train_objs():
    # parity with reference single_gpu.py:48-52
    train_set = ToyDataset(2048)
    model = toy_model(20, 1)
    optimizer = FusedSGD(model.parameters(), lr=1e-3)
    return train_set, model, optimizer


def main(device, total_epochs: int, save_every: int):
    dataset, model, optimizer = load_train_objs()
    train_data = prepare_dataloader(dataset, batch_size=32)
    trainer = Trainer(model, train_data, optimizer, device, save_every,
                      wrap_ddp=False,
                      engine=os.environ.get("MI355X_ENGINE", "hooks"))
    trainer.train(total_epochs)


if __name__ == "__main__":
    total_epochs = int(sys.argv[1])
    save_every = int(sys.argv[2])
    device = 0 if torch.cuda.is_available() else "cpu"
    main(device, total_epochs, save_every)
