"""Tensor-parallel linear layers over RCCL/xGMI.

Parity target: Apex/Megatron Column/RowParallelLinear as used by the
reference's NeMo path (SURVEY.md §2.2 TP row; modeling_nemo_ppo.py:93-120).

Math (Megatron scheme):
- ColumnParallelLinear: W split by OUTPUT rows; y_local = x @ W_l^T.  The
  backward all-reduces dx over the TP group (``copy_to_tp`` region on x).
- RowParallelLinear: W split by INPUT columns; y = all_reduce_tp(x_l @ W_l^T)
  (``reduce_from_tp`` region), bias added after the reduce.

Replicated modules (embeddings, norms, lm_head) need no extra grad sync:
column-parallel backward restores full input grads, so replicated weights see
identical gradients on every TP rank by construction.
"""

from typing import Optional

import torch
import torch.distributed as dist
import torch.nn as nn

from . import topo


class _CopyToTP(torch.autograd.Function):
    """Identity forward; all-reduce gradient over the TP group in backward."""

    @staticmethod
    def forward(ctx, x):
        return x

    @staticmethod
    def backward(ctx, grad):
        if topo.tp_size() > 1:
            grad = grad.contiguous()
            dist.all_reduce(grad, dist.ReduceOp.SUM, group=topo.tp_group())
        return grad


class _ReduceFromTP(torch.autograd.Function):
    """All-reduce forward over the TP group; identity backward."""

    @staticmethod
    def forward(ctx, x):
        if topo.tp_size() > 1:
            x = x.contiguous()
            dist.all_reduce(x, dist.ReduceOp.SUM, group=topo.tp_group())
        return x

    @staticmethod
    def backward(ctx, grad):
        return grad


def copy_to_tp(x: torch.Tensor) -> torch.Tensor:
    return _CopyToTP.apply(x)


def reduce_from_tp(x: torch.Tensor) -> torch.Tensor:
    return _ReduceFromTP.apply(x)


# --- sequence-parallel region functions (Megatron SP; SURVEY.md §2.2 SP row) ----


def _sp_all_gather(x: torch.Tensor) -> torch.Tensor:
    """Concatenate sequence shards along dim 1 from the TP group."""
    W = topo.tp_size()
    outs = [torch.empty_like(x) for _ in range(W)]
    dist.all_gather(outs, x.contiguous(), group=topo.tp_group())
    return torch.cat(outs, dim=1)


def _sp_reduce_scatter(x: torch.Tensor) -> torch.Tensor:
    """Sum-reduce along the TP group and keep this rank's sequence shard."""
    W = topo.tp_size()
    x = x.contiguous()
    dist.all_reduce(x, dist.ReduceOp.SUM, group=topo.tp_group())
    T = x.shape[1] // W
    r = topo.tp_rank()
    return x[:, r * T : (r + 1) * T].contiguous()


class _GatherFromSP(torch.autograd.Function):
    """fwd: all-gather sequence shards; bwd: reduce-scatter gradients
    (Megatron gather_from_sequence_parallel_region)."""

    @staticmethod
    def forward(ctx, x):
        return _sp_all_gather(x)

    @staticmethod
    def backward(ctx, grad):
        return _sp_reduce_scatter(grad)


class _ReduceScatterToSP(torch.autograd.Function):
    """fwd: reduce-scatter to sequence shards; bwd: all-gather gradients."""

    @staticmethod
    def forward(ctx, x):
        return _sp_reduce_scatter(x)

    @staticmethod
    def backward(ctx, grad):
        return _sp_all_gather(grad)


class _ScatterToSP(torch.autograd.Function):
    """fwd: keep this rank's sequence shard; bwd: all-gather gradients."""

    @staticmethod
    def forward(ctx, x):
        W = topo.tp_size()
        T = x.shape[1] // W
        r = topo.tp_rank()
        return x[:, r * T : (r + 1) * T].contiguous()

    @staticmethod
    def backward(ctx, grad):
        return _sp_all_gather(grad)


class _GatherFromSPReplicated(torch.autograd.Function):
    """fwd: all-gather sequence shards; bwd: take THIS rank's shard of the
    gradient.  Correct when the downstream computation (loss) is REPLICATED
    identically on every TP rank — the framework's convention for the final
    lm_head/loss region (reduce-scatter there would double-count the
    replicated gradients)."""

    @staticmethod
    def forward(ctx, x):
        return _sp_all_gather(x)

    @staticmethod
    def backward(ctx, grad):
        W = topo.tp_size()
        T = grad.shape[1] // W
        r = topo.tp_rank()
        return grad[:, r * T : (r + 1) * T].contiguous()


def gather_from_sp(x):
    return _GatherFromSP.apply(x)


def gather_from_sp_replicated(x):
    return _GatherFromSPReplicated.apply(x)


def reduce_scatter_to_sp(x):
    return _ReduceScatterToSP.apply(x)


def scatter_to_sp(x):
    return _ScatterToSP.apply(x)


class ColumnParallelLinear(nn.Module):
    """y_local = x @ W_l^T + b_l with W sharded on output dim.

    With ``sequence_parallel`` the input arrives sequence-sharded and is
    all-gathered here (grad reduce-scatters back) — the SP entry point."""

    def __init__(self, in_features: int, out_features: int, bias: bool = True):
        super().__init__()
        tp = topo.tp_size()
        assert out_features % tp == 0, f"out_features {out_features} % tp {tp}"
        self.in_features = in_features
        self.out_features = out_features
        self.out_local = out_features // tp
        self.weight = nn.Parameter(torch.empty(self.out_local, in_features))
        self.bias = nn.Parameter(torch.zeros(self.out_local)) if bias else None
        self.sequence_parallel = False

    def forward(self, x):
        if self.sequence_parallel and topo.tp_size() > 1:
            x = gather_from_sp(x)
        else:
            x = copy_to_tp(x)
        return torch.nn.functional.linear(x, self.weight, self.bias)


class RowParallelLinear(nn.Module):
    """y = all_reduce(x_l @ W_l^T) + b with W sharded on input dim.

    With ``sequence_parallel`` the all-reduce becomes a reduce-scatter to
    sequence shards — the SP exit point."""

    def __init__(self, in_features: int, out_features: int, bias: bool = True):
        super().__init__()
        tp = topo.tp_size()
        assert in_features % tp == 0, f"in_features {in_features} % tp {tp}"
        self.in_features = in_features
        self.out_features = out_features
        self.in_local = in_features // tp
        self.weight = nn.Parameter(torch.empty(out_features, self.in_local))
        self.bias = nn.Parameter(torch.zeros(out_features)) if bias else None
        self.sequence_parallel = False

    def forward(self, x):
        y = torch.nn.functional.linear(x, self.weight, None)
        if self.sequence_parallel and topo.tp_size() > 1:
            y = reduce_scatter_to_sp(y)
        else:
            y = reduce_from_tp(y)
        if self.bias is not None:
            y = y + self.bias
        return y


# ---------------------------------------------------------------------------
# checkpoint sharding: full state dict -> this TP rank's shard
# ---------------------------------------------------------------------------


def shard_qkv_weight(w: torch.Tensor, num_heads: int, num_kv_heads: int, head_dim: int,
                     tp_rank: int, tp: int) -> torch.Tensor:
    """Split a fused [(Hq+2Hkv)*D, in] qkv weight (or [.,] bias) head-aligned:
    this rank keeps its Hq/tp query heads + Hkv/tp k heads + Hkv/tp v heads."""
    qd = num_heads * head_dim
    kd = num_kv_heads * head_dim
    q, k, v = w[:qd], w[qd : qd + kd], w[qd + kd :]
    hq_l = num_heads // tp * head_dim
    hk_l = num_kv_heads // tp * head_dim
    return torch.cat([
        q[tp_rank * hq_l : (tp_rank + 1) * hq_l],
        k[tp_rank * hk_l : (tp_rank + 1) * hk_l],
        v[tp_rank * hk_l : (tp_rank + 1) * hk_l],
    ], dim=0)


def shard_gate_up_weight(w: torch.Tensor, intermediate: int, tp_rank: int, tp: int) -> torch.Tensor:
    """Split a fused [2i, in] SwiGLU gate_up weight keeping [gate_l ; up_l]."""
    gate, up = w[:intermediate], w[intermediate:]
    i_l = intermediate // tp
    return torch.cat([
        gate[tp_rank * i_l : (tp_rank + 1) * i_l],
        up[tp_rank * i_l : (tp_rank + 1) * i_l],
    ], dim=0)


def shard_state_dict_tp(sd: dict, cfg, tp_rank: int, tp: int) -> dict:
    """Shard a full native state dict for this TP rank (load-time resharding;
    the NeMo path's mp_rank_XX analog, SURVEY.md §5 checkpoint notes)."""
    if tp <= 1:
        return sd
    out = {}
    i = cfg.intermediate_size
    for k, v in sd.items():
        if ".attn.qkv_proj." in k:
            out[k] = shard_qkv_weight(v, cfg.num_heads, cfg.num_kv_heads, cfg.head_dim, tp_rank, tp)
        elif ".attn.o_proj.weight" in k:
            cols = v.shape[1] // tp
            out[k] = v[:, tp_rank * cols : (tp_rank + 1) * cols].contiguous()
        elif ".mlp.gate_up_proj." in k:
            out[k] = shard_gate_up_weight(v, i, tp_rank, tp)
        elif ".mlp.fc_in." in k:
            rows = v.shape[0] // tp
            out[k] = v[rows * tp_rank : rows * (tp_rank + 1)].contiguous()
        elif ".mlp.down_proj.weight" in k:
            cols = v.shape[1] // tp
            out[k] = v[:, tp_rank * cols : (tp_rank + 1) * cols].contiguous()
        elif ".mlp.down_proj.bias" in k or ".attn.o_proj.bias" in k:
            out[k] = v  # row-parallel bias is replicated (added post-reduce)
        else:
            out[k] = v
    return out


def merge_state_dicts_tp(shards: list, cfg, tp: int) -> dict:
    """Reassemble a full state dict from per-TP-rank shards (inverse of
    shard_state_dict_tp; accepts wrapper-prefixed keys)."""
    import torch

    out = {}
    for k in shards[0]:
        vs = [sd[k] for sd in shards]
        if ".attn.qkv_proj." in k:
            qd_l = cfg.num_heads // tp * cfg.head_dim
            kd_l = cfg.num_kv_heads // tp * cfg.head_dim
            qs = [v[:qd_l] for v in vs]
            ks = [v[qd_l : qd_l + kd_l] for v in vs]
            vvs = [v[qd_l + kd_l :] for v in vs]
            out[k] = torch.cat(qs + ks + vvs, dim=0)
        elif ".mlp.gate_up_proj." in k:
            i_l = cfg.intermediate_size // tp
            gates = [v[:i_l] for v in vs]
            ups = [v[i_l:] for v in vs]
            out[k] = torch.cat(gates + ups, dim=0)
        elif ".attn.o_proj.weight" in k or ".mlp.down_proj.weight" in k:
            out[k] = torch.cat(vs, dim=1)
        elif ".mlp.fc_in." in k:
            out[k] = torch.cat(vs, dim=0)
        else:
            out[k] = vs[0]
    return out
