"""Arena-based fused AdamW.

Every trainable parameter is repointed into one contiguous arena per dtype:
- param arena (model dtype, e.g. bf16) — ``p.data`` becomes a view;
- fp32 master arena (mixed-precision master weights);
- fp32 m/v moment arenas;
- grad arena (model dtype) — ``p.grad`` is pre-pinned as a view so autograd
  accumulates straight into it (gradient_accumulation_fusion, K10).

The whole optimizer step is then ONE HIP kernel launch per arena (vs. one per
tensor), the DP gradient averaging (1/N) is fused into the step, and the
bucketed gradient all-reduce (ddp.py) operates on contiguous arena slices.
"""

import math
from typing import Dict, Iterable, List, Optional, Tuple

import torch

from .. import ops
from ..ops import reference


class FusedAdamW(torch.optim.Optimizer):
    """AdamW over flat parameter arenas with fp32 masters.

    On GPU the step is the trlx_amd._C.fused_adamw kernel; on CPU a flat
    torch implementation with identical math (numerics tests compare them).
    """

    def __init__(self, params: Iterable[torch.nn.Parameter], lr: float = 1e-3,
                 betas: Tuple[float, float] = (0.9, 0.999), eps: float = 1e-8,
                 weight_decay: float = 0.0, grad_scale: float = 1.0,
                 zero: bool = False, zero_group=None):
        """``zero=True`` shards optimizer state over the DP group (ZeRO-1/2
        pattern): grads are reduce-scattered into this rank's
        arena shard, AdamW runs on the shard only (1/N of master/m/v memory
        AND step compute), and updated params are all-gathered — all on
        contiguous arena slices over RCCL."""
        defaults = dict(lr=lr, betas=betas, eps=eps, weight_decay=weight_decay)
        super().__init__(params, defaults)
        self.grad_scale = grad_scale
        self._step = 0
        import torch.distributed as dist

        self.zero = zero and dist.is_initialized() and dist.get_world_size(zero_group) > 1
        self.zero_group = zero_group
        self._zero_world = dist.get_world_size(zero_group) if self.zero else 1
        self._zero_rank = dist.get_rank(zero_group) if self.zero else 0
        self._arenas: List[Dict] = []
        self._build_arenas()

    def _build_arenas(self):
        """Repoint each param group's tensors into contiguous arenas — one
        arena per (device, dtype) within the group (bf16 trunk and fp32 heads
        coexist in one group).  Under ZeRO, arenas are padded to a multiple of
        the group size and master/m/v cover only this rank's shard."""
        W = self._zero_world
        for group in self.param_groups:
            params = [p for p in group["params"] if p.requires_grad]
            group_arenas = []
            by_kind = {}
            for p in params:
                by_kind.setdefault((p.device, p.dtype), []).append(p)
            for (device, dtype), plist in by_kind.items():
                total = sum(p.numel() for p in plist)
                padded = (total + W - 1) // W * W
                flat_p = torch.zeros(padded, device=device, dtype=dtype)
                flat_g = torch.zeros(padded, device=device, dtype=dtype)
                offset = 0
                offsets = []
                for p in plist:
                    n = p.numel()
                    flat_p[offset : offset + n].copy_(p.data.reshape(-1))
                    p.data = flat_p[offset : offset + n].view(p.shape)
                    p.grad = flat_g[offset : offset + n].view(p.shape)
                    offsets.append((offset, n))
                    offset += n
                shard = padded // W
                lo = self._zero_rank * shard
                p_shard = flat_p[lo : lo + shard] if self.zero else flat_p
                master = p_shard.float() if (dtype != torch.float32 or self.zero) else flat_p
                if self.zero and dtype == torch.float32:
                    master = p_shard.clone()
                m = torch.zeros_like(master)
                v = torch.zeros_like(master)
                group_arenas.append(dict(
                    params=plist, flat_p=flat_p, flat_g=flat_g, master=master, m=m, v=v,
                    offsets=offsets, shard=shard, lo=lo,
                ))
            self._arenas.append(group_arenas)

    # --- optimizer protocol -------------------------------------------------

    @torch.no_grad()
    def step(self, closure=None):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()
        self._step += 1
        import torch.distributed as dist

        for group, group_arenas in zip(self.param_groups, self._arenas):
            lr = group["lr"]
            beta1, beta2 = group["betas"]
            eps = group["eps"]
            wd = group["weight_decay"]
            for arena in group_arenas:
                if self.zero:
                    lo, shard = arena["lo"], arena["shard"]
                    p_sh = arena["flat_p"][lo : lo + shard]
                    g_sh = arena["flat_g"][lo : lo + shard]
                else:
                    p_sh, g_sh = arena["flat_p"], arena["flat_g"]
                if arena["flat_p"].is_cuda:
                    ext = ops._require_ext("fused_adamw")
                    ext.fused_adamw(p_sh, arena["master"], g_sh, arena["m"], arena["v"],
                                    self._step, lr, beta1, beta2, eps, wd, self.grad_scale)
                else:
                    self._cpu_step_flat(p_sh, g_sh, arena, lr, beta1, beta2, eps, wd)
                if self.zero:
                    if dist.get_backend(self.zero_group) == "gloo":
                        chunks = list(arena["flat_p"].chunk(self._zero_world))
                        dist.all_gather(chunks, p_sh.contiguous(), group=self.zero_group)
                    else:
                        dist.all_gather_into_tensor(arena["flat_p"], p_sh, group=self.zero_group)
        return loss

    def _cpu_step_flat(self, p_flat, g_flat, arena, lr, beta1, beta2, eps, wd):
        g = g_flat.float() * self.grad_scale
        m, v, master = arena["m"], arena["v"], arena["master"]
        m.mul_(beta1).add_(g, alpha=1 - beta1)
        v.mul_(beta2).addcmul_(g, g, value=1 - beta2)
        bc1 = 1 - beta1**self._step
        bc2 = 1 - beta2**self._step
        master.mul_(1 - lr * wd)
        denom = (v / bc2).sqrt().add_(eps)
        master.addcdiv_(m / bc1, denom, value=-lr)
        if not (p_flat.dtype == torch.float32 and p_flat.data_ptr() == master.data_ptr()):
            p_flat.copy_(master.to(p_flat.dtype))

    def zero_grad(self, set_to_none: bool = False):
        # grads are pre-pinned arena views: zero in place, never detach
        for group_arenas in self._arenas:
            for arena in group_arenas:
                arena["flat_g"].zero_()

    # --- persistence ----------------------------------------------------------

    def state_dict(self):
        return {
            "step": self._step,
            "grad_scale": self.grad_scale,
            "param_groups": [
                {k: v for k, v in g.items() if k != "params"} for g in self.param_groups
            ],
            "arenas": [
                [{"master": a["master"], "m": a["m"], "v": a["v"]} for a in group_arenas]
                for group_arenas in self._arenas
            ],
        }

    def load_state_dict(self, state):
        self._step = state["step"]
        self.grad_scale = state.get("grad_scale", self.grad_scale)
        for g, sg in zip(self.param_groups, state["param_groups"]):
            g.update(sg)
        for group_arenas, sga in zip(self._arenas, state["arenas"]):
            for arena, sa in zip(group_arenas, sga):
                arena["master"].copy_(sa["master"].to(arena["master"].device))
                arena["m"].copy_(sa["m"].to(arena["m"].device))
                arena["v"].copy_(sa["v"].to(arena["v"].device))
                if arena["flat_p"].dtype != torch.float32:
                    arena["flat_p"].copy_(arena["master"].to(arena["flat_p"].dtype))

    # --- arena access for the gradient reducer --------------------------------

    def grad_arenas(self) -> List[Tuple[torch.Tensor, List[Tuple[torch.nn.Parameter, int, int]]]]:
        """[(flat_grad, [(param, offset, numel), ...]), ...] for bucketing."""
        out = []
        for group_arenas in self._arenas:
            for arena in group_arenas:
                entries = [(p, off, n) for p, (off, n) in zip(arena["params"], arena["offsets"])]
                out.append((arena["flat_g"], entries))
        return out


class NullOptimizer:
    """Stand-in for ranks with NO trainable parameters (e.g. a fully frozen
    pipeline stage when ``num_layers_unfrozen`` keeps all trainable layers on
    the last stage).  step/zero_grad are no-ops; pairs with NullScheduler."""

    def __init__(self):
        self.param_groups = []

    def step(self, closure=None):
        pass

    def zero_grad(self, set_to_none: bool = True):
        pass

    def state_dict(self):
        return {}

    def load_state_dict(self, state):
        pass


class NullScheduler:
    """LR scheduler stand-in paired with NullOptimizer."""

    def step(self):
        pass

    def get_last_lr(self):
        return []

    def state_dict(self):
        return {}

    def load_state_dict(self, state):
        pass


def build_optimizer(model: torch.nn.Module, name: str, kwargs: dict, world: int = 1,
                    zero: bool = False):
    """Construct the optimizer named in the config (reference
    utils/__init__.py get_optimizer_class registry)."""
    params = [p for p in model.parameters() if p.requires_grad]
    if not params:
        return NullOptimizer()
    kwargs = dict(kwargs)
    if name in ("fused_adamw", "adamw"):
        from . import topo

        return FusedAdamW(params, grad_scale=1.0 / world, zero=zero,
                          zero_group=topo.dp_group(), **kwargs)
    if name == "adam":
        return torch.optim.Adam(params, **kwargs)
    if name == "sgd":
        return torch.optim.SGD(params, **kwargs)
    raise ValueError(f"Unknown optimizer: {name}")
