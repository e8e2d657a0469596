"""Native PEFT adapters (parity for the reference's peft usage:
tests/test_peft.py exercises LORA, PROMPT_TUNING and PREFIX_TUNING).

The peft package is not available offline, so the adapters are built in:

- ``LORA``: wraps target linears with y = Wx + (alpha/r) * B(A(dropout(x))).
- ``PROMPT_TUNING``: ``num_virtual_tokens`` trainable embeddings inserted
  between each row's left pads and its real tokens (the pad-aware analog of
  peft's prepend — left-pad prefixes stay contiguous so the causal/flash
  kernels' key_starts semantics hold; see transformer.insert_virtual_rows).
- ``PREFIX_TUNING``: per-layer trainable K/V pairs written into the same
  virtual slots after RoPE (the trunk never computes them).

Config dicts mirror peft's: ``{"peft_type": "LORA", "r": 8, ...}``,
``{"peft_type": "PROMPT_TUNING", "num_virtual_tokens": 8}``,
``{"peft_type": "PREFIX_TUNING", "num_virtual_tokens": 8}``.

``adapter_state_dict`` / save-load give the reference's adapter-checkpoint
behavior (modeling_base.py:183-241), and ``adapters_disabled()`` is the
disable_adapter context the PPO hydra path uses for reference logits under
peft (reference accelerate_ppo_trainer.py:74-77).
"""

import contextlib
import math
from typing import Dict, List, Optional

import torch
import torch.nn as nn

DEFAULT_TARGETS = ["qkv_proj", "o_proj", "fc_in", "gate_up_proj", "down_proj"]


class LoRALinear(nn.Module):
    """Wraps an existing linear-like module with a low-rank residual path."""

    def __init__(self, base: nn.Module, r: int, alpha: float, dropout: float = 0.0):
        super().__init__()
        self.base = base
        in_features = base.weight.shape[1]
        out_features = base.weight.shape[0]
        dtype = base.weight.dtype
        device = base.weight.device
        self.r = r
        self.scaling = alpha / r
        self.lora_A = nn.Parameter(torch.zeros(r, in_features, dtype=dtype, device=device))
        self.lora_B = nn.Parameter(torch.zeros(out_features, r, dtype=dtype, device=device))
        nn.init.kaiming_uniform_(self.lora_A, a=math.sqrt(5))
        self.dropout = nn.Dropout(dropout) if dropout > 0 else None
        self.enabled = True
        for p in self.base.parameters():
            p.requires_grad_(False)

    def forward(self, x):
        y = self.base(x)
        if not self.enabled:
            return y
        xa = self.dropout(x) if self.dropout is not None else x
        lora = torch.nn.functional.linear(
            torch.nn.functional.linear(xa.to(self.lora_A.dtype), self.lora_A), self.lora_B
        )
        return y + self.scaling * lora

    @property
    def weight(self):
        return self.base.weight


def _normalize_config(peft_config) -> Dict:
    if hasattr(peft_config, "to_dict"):
        peft_config = peft_config.to_dict()
    cfg = dict(peft_config)
    ptype = str(cfg.get("peft_type", "LORA")).upper()
    cfg["peft_type"] = ptype
    if ptype not in ("LORA", "PROMPT_TUNING", "PREFIX_TUNING"):
        raise NotImplementedError(
            f"peft_type {ptype} is not supported natively (LORA / PROMPT_TUNING / "
            f"PREFIX_TUNING)")
    return cfg


def apply_lora(model: nn.Module, peft_config) -> nn.Module:
    """Freeze the base model and inject LoRA into the target modules."""
    cfg = _normalize_config(peft_config)
    if cfg["peft_type"] != "LORA":
        return apply_peft(model, cfg)
    r = int(cfg.get("r", 8))
    alpha = float(cfg.get("lora_alpha", 2 * r))
    dropout = float(cfg.get("lora_dropout", 0.0))
    targets = cfg.get("target_modules") or DEFAULT_TARGETS

    for p in model.parameters():
        p.requires_grad_(False)

    replaced = 0
    for parent_name, parent in list(model.named_modules()):
        for child_name, child in list(parent.named_children()):
            if child_name in targets and hasattr(child, "weight") and child.weight.dim() == 2:
                setattr(parent, child_name, LoRALinear(child, r, alpha, dropout))
                replaced += 1
    if replaced == 0:
        raise ValueError(f"LoRA: no target modules matched {targets}")
    return model


def lora_parameters(model: nn.Module):
    for name, p in model.named_parameters():
        if "lora_A" in name or "lora_B" in name:
            yield name, p


def lora_state_dict(model: nn.Module) -> Dict[str, torch.Tensor]:
    return {name: p.detach().cpu() for name, p in lora_parameters(model)}


def load_lora_state_dict(model: nn.Module, sd: Dict[str, torch.Tensor]):
    own = dict(model.named_parameters())
    for name, val in sd.items():
        if name in own:
            own[name].data.copy_(val.to(own[name].device, own[name].dtype))


@contextlib.contextmanager
def lora_disabled(model: nn.Module):
    """Temporarily bypass all adapters (the reference-model forward under
    peft — peft's disable_adapter analog)."""
    layers = [m for m in model.modules() if isinstance(m, LoRALinear)]
    for layer in layers:
        layer.enabled = False
    try:
        yield
    finally:
        for layer in layers:
            layer.enabled = True


def has_lora(model: nn.Module) -> bool:
    return any(isinstance(m, LoRALinear) for m in model.modules())


class SoftPrompt(nn.Module):
    """PROMPT_TUNING adapter: ``n`` trainable virtual-token embeddings."""

    def __init__(self, n: int, hidden: int, dtype=None, device=None):
        super().__init__()
        self.embeddings = nn.Parameter(torch.randn(n, hidden, dtype=dtype, device=device) * 0.02)

    @property
    def n(self):
        return self.embeddings.shape[0]


class PrefixKV(nn.Module):
    """PREFIX_TUNING adapter: per-layer trainable K/V written into the
    virtual slots after RoPE (the trunk never computes these positions'
    attention keys/values)."""

    def __init__(self, num_layers: int, n: int, kv_heads: int, head_dim: int,
                 dtype=None, device=None):
        super().__init__()
        self.prefix_k = nn.Parameter(
            torch.randn(num_layers, kv_heads, n, head_dim, dtype=dtype, device=device) * 0.02)
        self.prefix_v = nn.Parameter(
            torch.randn(num_layers, kv_heads, n, head_dim, dtype=dtype, device=device) * 0.02)

    @property
    def n(self):
        return self.prefix_k.shape[2]


def apply_peft(model: nn.Module, peft_config) -> nn.Module:
    """Dispatch by peft_type; freezes the base model in every case."""
    cfg = _normalize_config(peft_config)
    ptype = cfg["peft_type"]
    if ptype == "LORA":
        return apply_lora(model, cfg)
    if ptype in ("PROMPT_TUNING", "PREFIX_TUNING"):
        n = int(cfg.get("num_virtual_tokens", 8))
        for p in model.parameters():
            p.requires_grad_(False)
        mcfg = model.config
        p0 = next(model.parameters())
        if ptype == "PROMPT_TUNING":
            model.soft_prompt = SoftPrompt(n, mcfg.hidden_size, dtype=p0.dtype, device=p0.device)
        else:
            model.prefix_kv = PrefixKV(mcfg.num_layers, n, mcfg.num_kv_heads, mcfg.head_dim,
                                       dtype=p0.dtype, device=p0.device)
        model.num_virtual_tokens = n
        return model
    raise NotImplementedError(f"peft_type {ptype} is not supported natively")


def adapter_parameters(model: nn.Module):
    for name, p in model.named_parameters():
        if "lora_" in name or "soft_prompt." in name or "prefix_kv." in name:
            yield name, p


def adapter_state_dict(model: nn.Module) -> Dict[str, torch.Tensor]:
    return {name: p.detach().cpu() for name, p in adapter_parameters(model)}


def load_adapter_state_dict(model: nn.Module, sd: Dict[str, torch.Tensor]):
    own = dict(model.named_parameters())
    for name, val in sd.items():
        if name in own:
            own[name].data.copy_(val.to(own[name].device, own[name].dtype))


@contextlib.contextmanager
def adapters_disabled(model: nn.Module):
    """Bypass ALL adapters (LoRA paths, soft prompts, KV prefixes) — the
    reference-model forward under peft."""
    layers = [m for m in model.modules() if isinstance(m, LoRALinear)]
    for layer in layers:
        layer.enabled = False
    had_virtual = getattr(model, "num_virtual_tokens", 0)
    if had_virtual:
        model.num_virtual_tokens = 0
    try:
        yield
    finally:
        for layer in layers:
            layer.enabled = True
        if had_virtual:
            model.num_virtual_tokens = had_virtual


def has_adapter(model: nn.Module) -> bool:
    return (has_lora(model) or getattr(model, "soft_prompt", None) is not None
            or getattr(model, "prefix_kv", None) is not None)
