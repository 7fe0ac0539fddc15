"""Stage 5 — profiled DDP training of ResNet-50 (reference
multigpu_profile.py).

`python multigpu_profile.py` spawns one process per GPU, trains a locally
defined ResNet-50 (torchvision is absent here — mi355x_ddp.models.resnet)
on the synthetic image dataset for 3 epochs with torch.profiler
(roctracer-backed Kineto on ROCm) capturing schedule wait=1/warmup=1/
active=5 and exporting TensorBoard traces to ./log/resnet50/ per rank
(reference multigpu_profile.py:80-91). Rank 0 saves the DDP-wrapped
state_dict to model_ddp.pth (reference :76-78).

This stage exercises the multi-bucket reduction path: ~102 MB of fp32
gradients per step across ~5 buckets.
"""

import os
import sys

import torch
import torch.multiprocessing as mp
from torch.distributed import destroy_process_group

from mi355x_ddp.data import RandomImageDataset, prepare_dataloader
from mi355x_ddp.models import resnet50
from mi355x_ddp.parallel import FusedSGD, ddp_setup
from mi355x_ddp.trainer import Trainer


def _tiny_cnn():
    # CPU test stand-in (MI355X_PROFILE_MODEL=tiny): keeps the multi-bucket
    # reducer + profiler machinery exercised without ResNet-50's CPU cost.
    import torch.nn as nn
    return nn.Sequential(
        nn.Conv2d(3, 8, 3, stride=2, padding=1), nn.BatchNorm2d(8),
        nn.ReLU(), nn.AdaptiveAvgPool2d(1), nn.Flatten(),
        nn.Linear(8, 1000))


def load_train_objs(dataset_size: int = 2048):
    shape = (3, 224, 224)
    kind = os.environ.get("MI355X_PROFILE_MODEL", "resnet50")
    if kind == "tiny":
        shape = (3, 32, 32)
        model = _tiny_cnn()
    elif kind == "vit":
        from mi355x_ddp.models import vit_l_32
        model = vit_l_32()  # the reference imports it but leaves it
        # commented out (ref multigpu_profile.py:24); here it runs
    else:
        model = resnet50()
    train_set = RandomImageDataset(dataset_size, shape)
    optimizer = FusedSGD(model.parameters(), lr=1e-3)
    return train_set, model, optimizer


def main(rank: int, world_size: int, total_epochs: int = 3,
         profile: bool = True, dataset_size: int = 2048,
         batch_size: int = 32):
    ddp_setup(rank, world_size)
    try:
        dataset, model, optimizer = load_train_objs(dataset_size)
        train_data = prepare_dataloader(dataset, batch_size=batch_size,
                                        distributed=True,
                                        num_replicas=world_size, rank=rank)
        device = rank if torch.cuda.is_available() else "cpu"
        trainer = Trainer(model, train_data, optimizer, device,
                          save_every=10 ** 9,  # reference saves at the end
                          profile=profile, save_wrapped=True,
                          checkpoint_path="model_ddp.pth")
        trainer.train(total_epochs)
        if trainer.global_rank == 0:
            trainer.save_checkpoint()
    finally:
        destroy_process_group()


if __name__ == "__main__":
    total_epochs = int(sys.argv[1]) if len(sys.argv) > 1 else 3
    dataset_size = int(os.environ.get("MI355X_PROFILE_DATASET", 2048))
    batch_size = int(os.environ.get("MI355X_PROFILE_BATCH", 32))
    if os.environ.get("MI355X_FORCE_DEV0") == "1":
        # pre-flight rehearsal: N ranks time-sharing one device via IPC
        world_size = int(os.environ.get("MI355X_WORLD", 2))
    elif torch.cuda.is_available():
        world_size = torch.cuda.device_count()
    else:
        world_size = int(os.environ.get("MI355X_WORLD", 2))
    mp.spawn(main, args=(world_size, total_epochs, True, dataset_size,
                         batch_size), nprocs=world_size)
