"""Model-parallel topology: TP x DP process groups over the xGMI mesh.

Parity target: the reference's Apex ``parallel_state`` usage.
Rank layout: ``rank = dp_idx * tp_size + tp_idx`` — TP groups are CONTIGUOUS
ranks so tensor-parallel all-reduces stay on the densest xGMI paths within a
node (SURVEY.md §2.3 MI355X note), and the DP group strides across them.
"""

from typing import Optional

import torch.distributed as dist


class ParallelState:
    tp_size: int = 1
    pp_size: int = 1
    tp_group = None
    dp_group = None
    tp_rank: int = 0
    dp_rank: int = 0
    dp_size: int = 1
    initialized: bool = False


_STATE = ParallelState()


def state() -> ParallelState:
    return _STATE


def init_model_parallel(tp_size: int = 1, pp_size: int = 1) -> ParallelState:
    """Create TP/DP groups.  Safe to call with tp_size=1 (no-op topology)."""
    s = _STATE
    s.tp_size = tp_size
    s.pp_size = pp_size
    if not dist.is_initialized() or tp_size <= 1:
        s.tp_group = None
        s.dp_group = None
        s.tp_rank = 0
        s.dp_rank = dist.get_rank() if dist.is_initialized() else 0
        s.dp_size = dist.get_world_size() if dist.is_initialized() else 1
        s.initialized = True
        return s

    world = dist.get_world_size()
    rank = dist.get_rank()
    assert world % tp_size == 0, f"world {world} not divisible by tp {tp_size}"
    dp_size = world // tp_size

    # TP groups: contiguous rank blocks (intra-node xGMI locality)
    for dp_idx in range(dp_size):
        ranks = list(range(dp_idx * tp_size, (dp_idx + 1) * tp_size))
        g = dist.new_group(ranks)
        if rank in ranks:
            s.tp_group = g
            s.tp_rank = ranks.index(rank)
    # DP groups: same tp_idx across blocks
    for tp_idx in range(tp_size):
        ranks = list(range(tp_idx, world, tp_size))
        g = dist.new_group(ranks)
        if rank in ranks:
            s.dp_group = g
            s.dp_rank = ranks.index(rank)
    s.dp_size = dp_size
    s.initialized = True
    return s


def tp_size() -> int:
    return _STATE.tp_size


def tp_rank() -> int:
    return _STATE.tp_rank


def tp_group():
    return _STATE.tp_group


def dp_group():
    """The data-parallel group (None = default/world when TP is off)."""
    return _STATE.dp_group


def dp_rank() -> int:
    if _STATE.initialized:
        return _STATE.dp_rank
    return dist.get_rank() if dist.is_initialized() else 0


def dp_size() -> int:
    if _STATE.initialized:
        return _STATE.dp_size
    return dist.get_world_size() if dist.is_initialized() else 1


def reset():
    """Testing helper: forget the topology."""
    global _STATE
    _STATE = ParallelState()
