"""Drop-in module for user code written against the reference's `utils.py`
(`from utils import MyTrainDataset`, reference single_gpu.py:3 /
multigpu.py:3): re-exports this framework's synthetic datasets under the
reference's public names."""

from mi355x_ddp.data import (MyRandomDataset, MyTrainDataset,  # noqa: F401
                             RandomImageDataset, ToyDataset)
