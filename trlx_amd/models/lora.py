"""Native LoRA adapters (peft-parity for the reference's LORA path).

The peft package is not available offline, so low-rank adapters are built in:
``apply_lora(model, peft_config)`` freezes the base weights and wraps the
target linear layers with y = Wx + (alpha/r) * B(A(dropout(x))).  The config
dict mirrors peft's: ``{"peft_type": "LORA", "r": 8, "lora_alpha": 32,
"lora_dropout": 0.0, "target_modules": ["qkv_proj", "o_proj"]}``.

``lora_state_dict`` / adapter save-load give the reference's
adapter-checkpoint behavior (modeling_base.py:183-241), and
``lora_disabled()`` is the disable_adapter context the PPO hydra path uses
for reference logits under peft (reference accelerate_ppo_trainer.py:74-77:
no frozen branch when peft is active).
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
    if ptype != "LORA":
        raise NotImplementedError(f"peft_type {ptype} is not supported natively (LORA only)")
    return cfg


def apply_lora(model: nn.Module, peft_config) -> nn.Module:
    """Freeze the base model and inject LoRA into the target modules."""
    cfg = _normalize_config(peft_config)
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
