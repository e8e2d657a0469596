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
    pp_group = None
    tp_rank: int = 0
    dp_rank: int = 0
    pp_rank: int = 0
    pp_ranks = None
    dp_size: int = 1
    initialized: bool = False


_STATE = ParallelState()


def state() -> ParallelState:
    return _STATE


def init_model_parallel(tp_size: int = 1, pp_size: int = 1) -> ParallelState:
    """Create TP/PP/DP groups.  Safe to call with tp=pp=1 (no-op topology)."""
    s = _STATE
    s.tp_size = tp_size
    s.pp_size = pp_size
    if not dist.is_initialized() or (tp_size <= 1 and pp_size <= 1):
        s.tp_group = None
        s.dp_group = None
        s.tp_rank = 0
        s.dp_rank = dist.get_rank() if dist.is_initialized() else 0
        s.dp_size = dist.get_world_size() if dist.is_initialized() else 1
        s.initialized = True
        return s

    world = dist.get_world_size()
    rank = dist.get_rank()
    assert world % (tp_size * pp_size) == 0, \
        f"world {world} not divisible by tp*pp {tp_size * pp_size}"
    dp_size = world // (tp_size * pp_size)
    block = tp_size * pp_size  # ranks per model replica

    # TP groups: contiguous rank blocks (intra-node xGMI locality)
    s.tp_group = None
    s.tp_rank = 0
    if tp_size > 1:
        for b in range(world // tp_size):
            ranks = list(range(b * tp_size, (b + 1) * tp_size))
            g = dist.new_group(ranks)
            if rank in ranks:
                s.tp_group = g
                s.tp_rank = ranks.index(rank)
    # PP groups: stages tp_size apart within a replica block
    s.pp_group = None
    s.pp_rank = 0
    s.pp_ranks = None
    if pp_size > 1:
        for dp_idx in range(dp_size):
            for tp_idx in range(tp_size):
                ranks = [dp_idx * block + p * tp_size + tp_idx for p in range(pp_size)]
                g = dist.new_group(ranks)
                if rank in ranks:
                    s.pp_group = g
                    s.pp_rank = ranks.index(rank)
                    s.pp_ranks = ranks
    # DP groups: same (pp_idx, tp_idx) across replica blocks
    for off in range(block):
        ranks = list(range(off, world, block))
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


def pp_size() -> int:
    return _STATE.pp_size


def pp_group():
    return _STATE.pp_group


def pp_rank() -> int:
    return _STATE.pp_rank


def reset():
    """Testing helper: forget the topology."""
    global _STATE
    _STATE = ParallelState()
