"""General utilities.

Parity target: reference trlx/utils/__init__.py (set_seed, significant,
optimizer/scheduler registries, Clock, tree_map, infinite_dataloader).
"""

import math
import os
import random
import subprocess
import time
from enum import Enum
from numbers import Number
from typing import Any, Dict, Iterable, Optional, Tuple

import numpy as np
import torch
from torch.optim.lr_scheduler import CosineAnnealingLR, LinearLR


def print_rank_0(*message):
    """Print only on rank 0."""
    if int(os.environ.get("RANK", 0)) == 0:
        print(*message)


def significant(x: Number, ndigits=2) -> Number:
    """Cut the number up to its ``ndigits`` after the most significant."""
    if isinstance(x, torch.Tensor):
        x = x.item()
    if not isinstance(x, Number) or isinstance(x, bool) or x == 0:
        return x
    return round(x, ndigits - int(math.floor(math.log10(abs(x)))))


def set_seed(seed: int, rank_offset: Optional[int] = None):
    """Seed python/numpy/torch; offsets by rank so DP rollouts decorrelate
    (reference trlx/utils/__init__.py:44).  Under tensor parallelism pass
    ``rank_offset=dp_rank`` so TP peers share the stream."""
    if rank_offset is None:
        rank_offset = int(os.environ.get("RANK", 0))
    seed = int(seed) + rank_offset
    random.seed(seed)
    np.random.seed(seed)
    torch.manual_seed(seed)
    if torch.cuda.is_available():
        torch.cuda.manual_seed(seed)


def get_distributed_config() -> Dict[str, Any]:
    """Describe the distributed launch for run metadata."""
    return {
        "rank": int(os.environ.get("RANK", 0)),
        "local_rank": int(os.environ.get("LOCAL_RANK", 0)),
        "world_size": int(os.environ.get("WORLD_SIZE", 1)),
        "master_addr": os.environ.get("MASTER_ADDR"),
        "master_port": os.environ.get("MASTER_PORT"),
        "backend": "rccl" if torch.cuda.is_available() else "gloo",
    }


class OptimizerName(str, Enum):
    """Supported optimizer names."""

    ADAM = "adam"
    ADAMW = "adamw"
    FUSED_ADAMW = "fused_adamw"  # native arena AdamW (HIP multi-tensor kernel)
    SGD = "sgd"


def get_optimizer_class(name) -> type:
    """Map a name to a torch optimizer class (the native fused AdamW is
    constructed by the trainer directly — see trlx_amd/parallel/optim.py)."""
    name = OptimizerName(name) if not isinstance(name, OptimizerName) else name
    if name == OptimizerName.ADAM:
        return torch.optim.Adam
    if name in (OptimizerName.ADAMW, OptimizerName.FUSED_ADAMW):
        return torch.optim.AdamW
    if name == OptimizerName.SGD:
        return torch.optim.SGD
    supported = [o.value for o in OptimizerName]
    raise ValueError(f"`{name}` is not a supported optimizer. Supported: {supported}")


class SchedulerName(str, Enum):
    """Supported scheduler names."""

    COSINE_ANNEALING = "cosine_annealing"
    LINEAR = "linear"


def get_scheduler_class(name) -> type:
    name = SchedulerName(name) if not isinstance(name, SchedulerName) else name
    if name == SchedulerName.COSINE_ANNEALING:
        return CosineAnnealingLR
    if name == SchedulerName.LINEAR:
        return LinearLR
    supported = [s.value for s in SchedulerName]
    raise ValueError(f"`{name}` is not a supported scheduler. Supported: {supported}")


class Clock:
    """Throughput/ETA helper (parity: reference utils Clock)."""

    def __init__(self):
        self.start = time.time()
        self.total_time = 0
        self.total_samples = 0

    def tick(self, samples: int = 0) -> float:
        """Returns time (s) since last call to tick(). Also records samples."""
        end = time.time()
        delta = end - self.start
        self.start = end
        if samples != 0:
            self.total_time += delta
            self.total_samples += samples
        return delta

    def get_stat(self, n_samp: int = 1000, reset: bool = False):
        """Returns average time (s) per n_samp samples processed."""
        sec_per_samp = self.total_time / max(self.total_samples, 1)
        if reset:
            self.total_time = 0
            self.total_samples = 0
        return sec_per_samp * n_samp


def tree_map(f, tree: Any) -> Any:
    """Apply f to all leaves in a nested dict/list/tuple/dataclass of tensors."""
    if hasattr(tree, "__dataclass_fields__"):
        return tree.__class__(**{k: tree_map(f, v) for k, v in tree.__dict__.items()})
    if isinstance(tree, dict):
        return {k: tree_map(f, v) for k, v in tree.items()}
    if isinstance(tree, (list, tuple)):
        return tree.__class__(tree_map(f, v) for v in tree)
    return f(tree)


def to_device(tree, device, non_blocking: bool = False):
    """Move all tensors in a nested structure to ``device``."""
    return tree_map(
        lambda x: x.to(device, non_blocking=non_blocking) if isinstance(x, torch.Tensor) else x, tree
    )


def filter_non_scalars(xs: Dict) -> Dict:
    """Only keep scalar values (for tracker logging)."""
    ys = {}
    for k, v in xs.items():
        try:
            ys[k] = float(v)
        except (TypeError, ValueError):
            continue
    return ys


def get_git_tag() -> Tuple[str, str]:
    """Returns (commit hash, branch name) when in a git repo."""
    try:
        output = subprocess.check_output("git log --format='%h/%as' -n1".split(), stderr=subprocess.DEVNULL)
        branch = subprocess.check_output("git rev-parse --abbrev-ref HEAD".split(), stderr=subprocess.DEVNULL)
        return branch.decode()[:-1], output.decode()[1:-2]
    except (subprocess.CalledProcessError, FileNotFoundError):
        return "unknown", "unknown"


def infinite_dataloader(dataloader: Iterable, sampler=None) -> Iterable:
    """Cycle a dataloader forever, bumping the sampler epoch each wrap so
    distributed shuffles differ per epoch (reference utils/__init__.py:240-250)."""
    epoch = 0
    if sampler is None:
        sampler = getattr(dataloader, "sampler", None)
    while True:
        for batch in dataloader:
            yield batch
        epoch += 1
        if hasattr(sampler, "set_epoch"):
            sampler.set_epoch(epoch)
        if sampler is not None and hasattr(sampler, "set_epoch"):
            sampler.set_epoch(epoch)
