"""Data layer: synthetic datasets, loader factory, and a from-scratch
distributed sampler.

Capability parity targets:
- reference `utils.py:4-13`  -> ToyDataset (2048 x (rand(20), rand(1)))
- reference `utils.py:16-26` -> RandomImageDataset (lazy (rand(shape), rand(1000)))
- reference `prepare_dataloader` (`single_gpu.py:55-61`, `multigpu.py:72-79`)
- torch `DistributedSampler` (implicit; `multigpu.py:77-78`) re-implemented
  here as `ShardedSampler` so the framework owns the sharding contract.
"""

from __future__ import annotations

import math
from typing import Iterator, Optional, Sequence, Tuple

import torch
from torch.utils.data import DataLoader, Dataset


class ToyDataset(Dataset):
    """2048 samples of (rand(20), rand(1)), materialized eagerly.

    Mirrors reference utils.py:4-13 (`MyTrainDataset`). The whole dataset is
    stored as two contiguous tensors rather than a python list of pairs —
    this makes a single H2D upload of the full dataset possible (164 KB for
    the default size; trivially resident in 288 GB of HBM3E), which the fast
    device-resident path in `bench.py` and the Trainer exploit.
    """

    def __init__(self, size: int = 2048, in_features: int = 20,
                 out_features: int = 1, seed: Optional[int] = None):
        self.size = size
        gen = torch.Generator()
        if seed is not None:
            gen.manual_seed(seed)
        self.inputs = torch.rand(size, in_features, generator=gen)
        self.targets = torch.rand(size, out_features, generator=gen)

    def __len__(self) -> int:
        return self.size

    def __getitem__(self, index: int) -> Tuple[torch.Tensor, torch.Tensor]:
        return self.inputs[index], self.targets[index]


# Alias with the reference's public name so users of the reference find it.
MyTrainDataset = ToyDataset


class RandomImageDataset(Dataset):
    """Lazy synthetic image dataset: (rand(*input_shape), rand(*target_shape)).

    Mirrors reference utils.py:16-26 (`MyRandomDataset`); used by the
    profiling entrypoint with input_shape=(3, 224, 224) for ResNet-50.
    """

    def __init__(self, size: int, input_shape: Sequence[int],
                 target_shape: Sequence[int] = (1000,)):
        self.size = size
        self.shape = tuple(input_shape)
        self.target_shape = tuple(target_shape)

    def __len__(self) -> int:
        return self.size

    def __getitem__(self, index: int) -> Tuple[torch.Tensor, torch.Tensor]:
        return torch.rand(self.shape), torch.rand(self.target_shape)


MyRandomDataset = RandomImageDataset


class ShardedSampler(torch.utils.data.Sampler):
    """From-scratch replacement for torch's DistributedSampler.

    Contract (kept identical to torch's so loaders behave the same):
    - pads the index list by wrapping so every rank gets exactly
      ceil(len(dataset)/world) samples;
    - rank r takes indices[r::world] of the (optionally shuffled) list;
    - shuffle order is seeded by (seed + epoch) so `set_epoch` reshuffles —
      the reference never calls set_epoch (wart); our Trainer
      does call it every epoch.
    """

    def __init__(self, dataset: Dataset, num_replicas: Optional[int] = None,
                 rank: Optional[int] = None, shuffle: bool = True,
                 seed: int = 0, drop_last: bool = False):
        if num_replicas is None or rank is None:
            import torch.distributed as dist
            if dist.is_available() and dist.is_initialized():
                num_replicas = num_replicas or dist.get_world_size()
                rank = rank if rank is not None else dist.get_rank()
            else:
                num_replicas = num_replicas or 1
                rank = rank or 0
        if not (0 <= rank < num_replicas):
            raise ValueError(f"rank {rank} out of range for world {num_replicas}")
        self.dataset = dataset
        self.num_replicas = num_replicas
        self.rank = rank
        self.shuffle = shuffle
        self.seed = seed
        self.epoch = 0
        self.drop_last = drop_last
        n = len(dataset)
        if drop_last and n % num_replicas:
            self.num_samples = n // num_replicas
        else:
            self.num_samples = math.ceil(n / num_replicas)
        self.total_size = self.num_samples * num_replicas

    def set_epoch(self, epoch: int) -> None:
        self.epoch = epoch

    def __len__(self) -> int:
        return self.num_samples

    def __iter__(self) -> Iterator[int]:
        n = len(self.dataset)
        if self.shuffle:
            gen = torch.Generator()
            gen.manual_seed(self.seed + self.epoch)
            indices = torch.randperm(n, generator=gen).tolist()
        else:
            indices = list(range(n))
        if not self.drop_last:
            pad = self.total_size - len(indices)
            if pad > 0:
                # wrap-around padding, same as torch's sampler
                reps = math.ceil(pad / n)
                indices += (indices * reps)[:pad]
        else:
            indices = indices[: self.total_size]
        assert len(indices) == self.total_size
        shard = indices[self.rank :: self.num_replicas]
        assert len(shard) == self.num_samples
        return iter(shard)


class DevicePrefetcher:
    """Pinned-host -> device copy pipeline.

    The reference requests pinned loader memory but copies synchronously
    (`pin_memory=True` at single_gpu.py:59 yet `.to(device)` without
    `non_blocking`, multigpu.py:49-50 — a listed wart). This wraps a
    DataLoader so batch s+1 is uploaded on a DEDICATED copy stream while
    the model computes batch s: the compute stream only waits on the
    recorded copy event, never on the DMA itself. On CPU devices it is a
    transparent passthrough.
    """

    def __init__(self, loader, device: torch.device):
        self.loader = loader
        self.device = device
        self._stream = (torch.cuda.Stream(device=device)
                        if device.type == "cuda" else None)

    def __len__(self) -> int:
        return len(self.loader)

    def _upload(self, batch):
        with torch.cuda.stream(self._stream):
            moved = tuple(
                t.to(self.device, non_blocking=True) if torch.is_tensor(t)
                else t for t in batch)
            ev = torch.cuda.Event()
            ev.record(self._stream)
        return moved, ev

    def __iter__(self):
        if self._stream is None:
            yield from self.loader
            return
        pending = None
        main = torch.cuda.current_stream()
        for batch in self.loader:
            nxt = self._upload(batch)
            if pending is not None:
                moved, ev = pending
                main.wait_event(ev)
                for t in moved:
                    if torch.is_tensor(t):
                        t.record_stream(main)
                yield moved
            pending = nxt
        if pending is not None:
            moved, ev = pending
            main.wait_event(ev)
            for t in moved:
                if torch.is_tensor(t):
                    t.record_stream(main)
            yield moved


def prepare_dataloader(dataset: Dataset, batch_size: int,
                       distributed: bool = False,
                       shuffle: bool = True,
                       num_replicas: Optional[int] = None,
                       rank: Optional[int] = None,
                       seed: int = 0) -> DataLoader:
    """Loader factory mirroring the reference's prepare_dataloader
    (single_gpu.py:55-61 / multigpu.py:72-79).

    Distributed mode attaches a ShardedSampler (sharding + padding like the
    reference's DistributedSampler usage at multigpu.py:77-78). The
    reference passes shuffle=False to its sampler; we default to the
    sampler's epoch-seeded shuffle which the Trainer advances via set_epoch.
    pin_memory is only requested when a GPU is present (it is pointless and
    slow on a CPU-only host).
    """
    pin = torch.cuda.is_available()
    if distributed:
        sampler = ShardedSampler(dataset, num_replicas=num_replicas, rank=rank,
                                 shuffle=shuffle, seed=seed)
        return DataLoader(dataset, batch_size=batch_size, pin_memory=pin,
                          shuffle=False, sampler=sampler)
    return DataLoader(dataset, batch_size=batch_size, pin_memory=pin,
                      shuffle=shuffle)
