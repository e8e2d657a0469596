"""Pipeline framework: dataset/rollout-store bases + microbatch iterator.

Parity target: reference trlx/pipeline/__init__.py — the ``_DATAPIPELINE``
registry, BasePipeline (a Dataset with create_loader), BaseRolloutStore
(push / indexable / create_loader), and MiniBatchIterator (slices DataLoader
batches into num_mb microbatches for gradient accumulation; handles dict,
dataclass, and BatchEncoding batches).
"""

import sys
from abc import abstractmethod, abstractstaticmethod
from dataclasses import is_dataclass
from typing import Any, Callable, Dict, Iterable

from torch.utils.data import DataLoader, Dataset

from ..utils import logging

logger = logging.get_logger(__name__)

_DATAPIPELINE: Dict[str, type] = {}


def register_datapipeline(name):
    """Decorator registering a pipeline class by name."""

    def register_class(cls, name):
        _DATAPIPELINE[name] = cls
        setattr(sys.modules[__name__], name, cls)
        return cls

    if isinstance(name, str):
        name = name.lower()
        return lambda c: register_class(c, name)

    cls = name
    name = cls.__name__
    register_class(cls, name.lower())
    return cls


def dp_sampler(dataset, shuffle: bool = False, seed: int = 0):
    """A DistributedSampler over the DATA-parallel group, or None when
    single-process.

    Each DP rank iterates a disjoint shard per epoch (the reference shards via
    accelerate.prepare); TP/PP peers share a dp_rank and therefore see
    identical data.  Returned for ANY multi-process run (even dp_size == 1,
    where it degenerates to a deterministic full-coverage shuffler): model-
    parallel peers must draw identical batches, and the global torch RNG has
    diverged across ranks by init time (per-stage/per-shard weight init).
    Callers must pass ``shuffle=False`` to the DataLoader when a sampler is
    returned and bump ``sampler.set_epoch`` across epochs
    (``utils.infinite_dataloader`` does this).
    """
    import torch.distributed as dist

    from ..parallel import topo

    if not dist.is_initialized() or dist.get_world_size() <= 1:
        return None
    from torch.utils.data import DistributedSampler

    return DistributedSampler(
        dataset, num_replicas=topo.dp_size(), rank=topo.dp_rank(),
        shuffle=shuffle, seed=seed, drop_last=False,
    )


class BasePipeline(Dataset):
    def __init__(self, path: str = "dataset"):
        super().__init__()

    @abstractmethod
    def __getitem__(self, index: int):
        pass

    @abstractmethod
    def __len__(self) -> int:
        pass

    @abstractmethod
    def create_loader(self, batch_size: int, shuffle: bool, prep_fn: Callable = None,
                      num_workers: int = 0) -> DataLoader:
        pass


class BaseRolloutStore(Dataset):
    def __init__(self, capacity=-1):
        self.history: Iterable[Any] = None
        self.capacity = capacity

    @abstractmethod
    def push(self, exps: Iterable[Any]):
        """Push experiences to the store."""
        pass

    def __getitem__(self, index: int):
        return self.history[index]

    def __len__(self) -> int:
        return len(self.history)

    @abstractmethod
    def create_loader(self, batch_size: int, shuffle: bool, prep_fn: Callable = None,
                      num_workers: int = 0) -> DataLoader:
        pass


class MiniBatchIterator:
    """Slice each DataLoader batch into ``num_mb`` microbatches of ``mb_size``
    (reference trlx/pipeline/__init__.py:105-177)."""

    def __init__(self, data_loader, mb_size: int, num_mb: int):
        self.data_loader = data_loader
        self.data_loader_iter = iter(data_loader)
        self.mb_size = mb_size
        self.num_mb = num_mb

    def __iter__(self):
        return self

    def __next__(self):
        batch = next(self.data_loader_iter)
        if batch is None:
            logger.warning("Not enough samples to fill a minibatch")
            raise StopIteration

        minibatches = []
        for mbi in range(self.num_mb):
            batch_dict = batch.__dict__ if is_dataclass(batch) else batch
            sliced = {}
            ok = True
            for key, value in batch_dict.items():
                lo = mbi * self.mb_size
                hi = (mbi + 1) * self.mb_size
                piece = value[lo:hi]
                if self.num_mb > 1 and len(piece) == 0:
                    ok = False
                    break
                if self.num_mb > 1 and len(piece) < self.mb_size:
                    logger.warning("MiniBatchIterator produced a short microbatch")
                sliced[key] = piece
            if not ok or not sliced:
                break
            if is_dataclass(batch):
                minibatches.append(batch.__class__(**sliced))
            else:
                try:
                    from transformers import BatchEncoding

                    if isinstance(batch, BatchEncoding):
                        minibatches.append(BatchEncoding(sliced))
                        continue
                except ImportError:
                    pass
                minibatches.append(sliced)

        if not minibatches:
            raise StopIteration
        return minibatches
