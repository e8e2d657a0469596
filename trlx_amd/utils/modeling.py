"""Model/math utilities shared by trainers and losses.

Parity target: reference trlx/utils/modeling.py — make_head, freezers,
get_global_statistics, whiten, logprobs_of_labels, flatten_dict, gather_dict,
get_tensor_stats, RunningMoments.  The tensor math routes through
``trlx_amd.ops`` so the HIP kernels run on GPU.
"""

from typing import Dict, MutableMapping, Optional, Tuple, Union

import torch
import torch.distributed as dist
import torch.nn as nn

from .. import ops


def make_head(n_embd: int, out: int, dtype: torch.dtype = torch.float32) -> nn.Sequential:
    """Value/Q-head MLP: Linear(h->2h) -> ReLU -> Linear(2h->out)
    (reference trlx/utils/modeling.py:13-19)."""
    return nn.Sequential(
        nn.Linear(n_embd, n_embd * 2, dtype=dtype),
        nn.ReLU(),
        nn.Linear(n_embd * 2, out, dtype=dtype),
    )


def logprobs_of_labels(logits: torch.Tensor, labels: torch.Tensor) -> torch.Tensor:
    """Log-probs of labels — fused HIP kernel on GPU."""
    return ops.logprobs_of_labels(logits, labels)


def get_global_statistics(
    xs: torch.Tensor, mask: Optional[torch.Tensor] = None, group=None
) -> Tuple[torch.Tensor, torch.Tensor, int]:
    """Global mean/variance of ``xs`` across the process group — two
    all-reduces (reference trlx/utils/modeling.py:185-197)."""
    if mask is not None:
        xs = xs[mask.bool()]
    device = xs.device
    sum_and_count = torch.tensor([xs.sum(), xs.numel()], dtype=torch.float64, device=device)
    if dist.is_initialized():
        dist.all_reduce(sum_and_count, dist.ReduceOp.SUM, group=group)
    global_sum, count = sum_and_count
    global_mean = global_sum / count
    sum_var = torch.sum((xs.double() - global_mean) ** 2)
    if dist.is_initialized():
        dist.all_reduce(sum_var, dist.ReduceOp.SUM, group=group)
    global_var = sum_var / count
    return global_mean.float(), global_var.float(), int(count.item())


def whiten(xs: torch.Tensor, shift_mean: bool = True, distributed: bool = True, group=None) -> torch.Tensor:
    """Whiten values (reference trlx/utils/modeling.py:200-210); distributed
    statistics by default when a process group is up."""
    distributed = distributed and dist.is_initialized()
    return ops.whiten(xs, shift_mean=shift_mean, distributed=distributed, group=group)


def flatten_dict(d, parent_key: str = "", sep: str = "/") -> dict:
    """Flatten a nested dict for tracker logging."""
    items = []
    for k, v in d.items():
        new_key = parent_key + sep + k if parent_key else k
        if isinstance(v, MutableMapping):
            items.extend(flatten_dict(v, new_key, sep=sep).items())
        else:
            items.append((new_key, v))
    return dict(items)


def gather_dict(obj: Dict, grad_state=None) -> Dict:
    """Gather a dict of lists from all ranks into one dict of concatenated
    lists (reference trlx/utils/modeling.py:238-259)."""
    if not dist.is_initialized():
        return obj
    objs = [None] * dist.get_world_size()
    dist.all_gather_object(objs, obj)
    acc, *objs = objs
    for obj in objs:
        for k in obj:
            acc[k].extend(obj[k])
    return acc


def get_tensor_stats(xs: torch.Tensor, mask: torch.Tensor, n: int) -> Dict:
    """Mean/min/max/std of masked values (for PPO stats logging)."""
    if xs.numel() == 0:
        zero = torch.tensor(0.0, device=xs.device)
        return dict(mean=zero, min=zero, max=zero, std=zero)
    mean = (xs * mask).sum() / n
    minimum = torch.where(mask.bool(), xs, torch.full_like(xs, float("inf"))).min()
    maximum = torch.where(mask.bool(), xs, torch.full_like(xs, float("-inf"))).max()
    denom = torch.clamp(torch.as_tensor(n, dtype=torch.float32, device=xs.device) - 1, min=1)
    std = torch.sqrt(((xs - mean) * mask).pow(2).sum() / denom)
    return dict(mean=mean, min=minimum, max=maximum, std=std)


class RunningMoments:
    """Cross-rank running mean/std of rollout rewards
    (reference trlx/utils/modeling.py:275-307)."""

    def __init__(self):
        self.mean = 0.0
        self.std = 1.0
        self.var = 1.0
        self.count = 1e-24

    def update(self, xs: torch.Tensor, group=None) -> Tuple[float, float]:
        """Update from a batch; returns the batch's own (mean, std)."""
        if dist.is_initialized():
            xs_mean, xs_var, xs_count = get_global_statistics(xs, group=group)
            xs_mean = float(xs_mean)
            xs_var = float(xs_var)
        else:
            xs_count = xs.numel()
            xs_var, xs_mean = torch.var_mean(xs.float(), unbiased=False)
            xs_mean = float(xs_mean)
            xs_var = float(xs_var)

        delta = xs_mean - self.mean
        tot_count = self.count + xs_count

        new_sum = xs_var * xs_count
        old_sum = self.var * self.count + delta**2 * self.count * xs_count / tot_count
        tot_sum = old_sum + new_sum

        self.mean += delta * xs_count / tot_count
        self.var = tot_sum / tot_count
        self.std = (self.var * tot_count / max(tot_count - 1, 1)) ** 0.5
        self.count = tot_count

        return xs_mean, (xs_var * xs_count / max(xs_count - 1, 1)) ** 0.5


def freeze_bottom_causal_layers(model, num_layers_unfrozen: int = 0):
    """Freeze everything except the top ``num_layers_unfrozen`` decoder blocks
    (+ final norm + heads). -1 trains everything; 0 freezes all blocks."""
    layers = model.layers
    if num_layers_unfrozen == -1:
        return
    if num_layers_unfrozen == 0:
        frozen = list(layers)
    else:
        frozen = list(layers)[:-num_layers_unfrozen]
    # embeddings are below the first unfrozen layer -> frozen too
    for p in model.embed_parameters():
        p.requires_grad_(False)
    for block in frozen:
        for p in block.parameters():
            p.requires_grad_(False)


def freeze_bottom_seq2seq_layers(model, num_layers_unfrozen: int = 0):
    """Seq2seq freezing (reference trlx/utils/modeling.py
    freeze_bottom_seq2seq_layers): freeze shared embeddings, the whole
    encoder, and all but the top ``num_layers_unfrozen`` decoder blocks."""
    if num_layers_unfrozen == -1:
        return
    model.shared.weight.requires_grad_(False)
    for p in model.encoder_blocks.parameters():
        p.requires_grad_(False)
    for p in model.encoder_final_norm.parameters():
        p.requires_grad_(False)
    blocks = list(model.decoder_blocks)
    frozen = blocks if num_layers_unfrozen == 0 else blocks[:-num_layers_unfrozen]
    for block in frozen:
        for p in block.parameters():
            p.requires_grad_(False)


def union_dicts(*dicts) -> Dict:
    out = {}
    for d in dicts:
        out.update(d)
    return out
