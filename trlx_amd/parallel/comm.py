"""Communication layer: RCCL over xGMI (torch.distributed backend "nccl" on
ROCm), gloo on CPU.

One place for every collective the framework issues (SURVEY.md §2.3 site
inventory): init, pad+gather of rollouts, rank-0 scatter of scores, scalar
all-reduces, object gathers.  Bucketed gradient all-reduce lives in ddp.py.
"""

import os
from datetime import timedelta
from typing import Any, Dict, List, Optional

import torch
import torch.distributed as dist


def world_size() -> int:
    if dist.is_initialized():
        return dist.get_world_size()
    return int(os.environ.get("WORLD_SIZE", 1))


def rank() -> int:
    if dist.is_initialized():
        return dist.get_rank()
    return int(os.environ.get("RANK", 0))


def local_rank() -> int:
    return int(os.environ.get("LOCAL_RANK", 0))


def is_main_process() -> bool:
    return rank() == 0


def init_distributed(timeout_minutes: int = 30) -> bool:
    """Initialize the default process group (RCCL on GPU, gloo on CPU) when
    launched with torchrun; no-op for single process.  Returns whether a
    process group is active."""
    if dist.is_initialized():
        return True
    if int(os.environ.get("WORLD_SIZE", 1)) <= 1:
        return False
    backend = "nccl" if torch.cuda.is_available() else "gloo"
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29500")
    if torch.cuda.is_available():
        torch.cuda.set_device(local_rank())
    dist.init_process_group(backend=backend, timeout=timedelta(minutes=timeout_minutes))
    return True


def get_device() -> torch.device:
    if torch.cuda.is_available():
        return torch.device("cuda", local_rank())
    return torch.device("cpu")


def barrier():
    if dist.is_initialized():
        if torch.cuda.is_available():
            dist.barrier(device_ids=[local_rank()])
        else:
            dist.barrier()


def pad_across_processes(tensor: torch.Tensor, dim: int = 1, pad_index: int = 0,
                         pad_first: bool = False) -> torch.Tensor:
    """Pad ``tensor`` along ``dim`` to the max size across ranks
    (reference: accelerate pad_across_processes)."""
    if not dist.is_initialized():
        return tensor
    size = torch.tensor([tensor.shape[dim]], device=tensor.device, dtype=torch.long)
    if torch.cuda.is_available():
        dist.all_reduce(size, dist.ReduceOp.MAX)
        max_size = int(size.item())
    else:
        dist.all_reduce(size, dist.ReduceOp.MAX)
        max_size = int(size.item())
    if max_size == tensor.shape[dim]:
        return tensor
    pad_shape = list(tensor.shape)
    pad_shape[dim] = max_size - tensor.shape[dim]
    pad = torch.full(pad_shape, pad_index, device=tensor.device, dtype=tensor.dtype)
    if pad_first:
        return torch.cat([pad, tensor], dim=dim)
    return torch.cat([tensor, pad], dim=dim)


def gather(tensor: torch.Tensor, dim: int = 0) -> torch.Tensor:
    """All-gather and concatenate along ``dim`` (rollout/eval sample gather,
    reference accelerate_ppo_trainer.py:292-300)."""
    if not dist.is_initialized():
        return tensor
    ws = dist.get_world_size()
    outs = [torch.empty_like(tensor) for _ in range(ws)]
    dist.all_gather(outs, tensor.contiguous())
    return torch.cat(outs, dim=dim)


def gather_object(obj: Any) -> List[Any]:
    if not dist.is_initialized():
        return [obj]
    outs = [None] * dist.get_world_size()
    dist.all_gather_object(outs, obj)
    return outs


def broadcast_scalar(value: float, src: int = 0, device=None) -> float:
    if not dist.is_initialized():
        return value
    t = torch.tensor([value], device=device or get_device(), dtype=torch.float64)
    dist.broadcast(t, src)
    return float(t.item())


def scatter_rows(scores: Optional[List[torch.Tensor]], rows: int, width: int,
                 device) -> torch.Tensor:
    """Rank 0 scatters per-rank [rows, width] chunks; others receive
    (reference accelerate_ppo_trainer.py:336-338)."""
    if not dist.is_initialized():
        assert scores is not None
        return scores[0].clone().detach()
    out = torch.empty((rows, width), device=device)
    dist.scatter(out, scatter_list=scores if rank() == 0 else None, src=0)
    return out


def all_reduce_mean(value: torch.Tensor, group=None) -> torch.Tensor:
    if dist.is_initialized():
        dist.all_reduce(value, dist.ReduceOp.AVG if torch.cuda.is_available() else dist.ReduceOp.SUM, group=group)
        if not torch.cuda.is_available():
            value = value / dist.get_world_size(group)
    return value


def all_reduce_max_flag(flag: bool, device=None) -> bool:
    """MAX all-reduce of a boolean (save-best agreement,
    reference accelerate_base_trainer.py:626-628)."""
    if not dist.is_initialized():
        return flag
    t = torch.tensor(int(flag), device=device or get_device())
    dist.all_reduce(t, dist.ReduceOp.MAX)
    return bool(t.item())


def preflight(log=print) -> bool:
    """Communication-environment sanity check at job startup (bench/trainer):
    logs the topology and verifies one small all-reduce / broadcast /
    all-gather with value checks over the default group.  Cheap (three
    scalar-ish collectives) and loud: first contact with a fresh multi-GPU
    node fails HERE with a clear message instead of deep inside training.
    Returns True when the collectives verify."""
    if not dist.is_initialized():
        return True
    w = world_size()
    r = rank()
    dev = get_device()
    if r == 0:
        info = {
            "backend": dist.get_backend(),
            "world_size": w,
            "devices_visible": torch.cuda.device_count() if torch.cuda.is_available() else 0,
            "rccl_env": {k: v for k, v in os.environ.items()
                         if k.startswith(("NCCL_", "RCCL_", "HSA_"))},
        }
        log(f"[comm.preflight] {info}")
    t = torch.tensor([float(r + 1)], device=dev)
    dist.all_reduce(t)
    expect = w * (w + 1) / 2
    if abs(float(t.item()) - expect) > 1e-4:
        raise RuntimeError(f"[comm.preflight] all_reduce wrong: got {t.item()}, want {expect}")
    b = torch.tensor([42.5 if r == 0 else 0.0], device=dev)
    dist.broadcast(b, 0)
    if abs(float(b.item()) - 42.5) > 1e-6:
        raise RuntimeError(f"[comm.preflight] broadcast wrong: got {b.item()}")
    buf = [torch.zeros(1, device=dev) for _ in range(w)]
    dist.all_gather(buf, torch.tensor([float(r)], device=dev))
    got = [int(x.item()) for x in buf]
    if got != list(range(w)):
        raise RuntimeError(f"[comm.preflight] all_gather wrong: {got}")
    if r == 0:
        log(f"[comm.preflight] OK: all_reduce/broadcast/all_gather verified over {w} ranks")
    return True
