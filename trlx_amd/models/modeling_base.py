"""Base model wrapper.

Parity target: reference trlx/models/modeling_base.py (PreTrainedModelWrapper:
from_pretrained / from_config / save_pretrained with separately-prefixed head
weights).  The wrapped model here is the native CausalTransformer; checkpoints
are HF-format directories (config.json + model.safetensors with HF weight
names) so vanilla transformers can load the base model, with the wrapper's
head weights stored alongside under their ``v_head.`` / ``ilql_heads.`` /
``frozen_head.`` prefixes (reference modeling_ppo.py:354-368 key scheme).
"""

import json
import os
from typing import Any, Dict, List, Optional, Union

import torch
import torch.nn as nn

from ..utils import logging
from .nn.config import PRESETS, TransformerConfig, preset
from .nn.convert import config_to_hf, load_hf_dir, save_hf_dir, state_dict_to_hf
from .nn.transformer import CausalTransformer

logger = logging.get_logger(__name__)

WRAPPER_HEADS_NAME = "wrapper_heads.pt"


class PreTrainedModelWrapper(nn.Module):
    """A wrapper around the native transformer plus method-specific heads."""

    _auto_model_parent_class = CausalTransformer
    _supported_modules: List[str] = []  # head module names, e.g. ["v_head"]
    _supported_args: List[str] = []

    def __init__(self, base_model: CausalTransformer, **kwargs):
        super().__init__()
        self.base_model = base_model
        self.config = base_model.config

    # --- construction -------------------------------------------------------

    @classmethod
    def _split_kwargs(cls, kwargs: Dict[str, Any]):
        supported = {}
        unsupported = {}
        for k, v in kwargs.items():
            if k in cls._supported_args:
                supported[k] = v
            else:
                unsupported[k] = v
        return supported, unsupported

    @classmethod
    def from_config(cls, config: Union[TransformerConfig, str], **kwargs):
        """Build with randomly initialized weights."""
        if isinstance(config, str):
            config = preset(config)
        wrapped_kwargs, _ = cls._split_kwargs(kwargs)
        base = CausalTransformer(config)
        if wrapped_kwargs.get("peft_config") is not None:
            from .lora import apply_lora

            apply_lora(base, wrapped_kwargs["peft_config"])
        return cls(base, **wrapped_kwargs)

    @classmethod
    def from_pretrained(cls, pretrained_model_name_or_path: Union[str, CausalTransformer], *,
                        revision=None, **kwargs):
        """Load from a local HF-format directory (or wrap an existing model).

        With no network, a bare model name (e.g. "gpt2") falls back to a
        random-init preset of the same architecture with a warning.
        """
        wrapped_kwargs, _ = cls._split_kwargs(kwargs)
        heads_sd = {}
        if isinstance(pretrained_model_name_or_path, CausalTransformer):
            base = pretrained_model_name_or_path
        elif isinstance(pretrained_model_name_or_path, PreTrainedModelWrapper):
            base = pretrained_model_name_or_path.base_model
        elif os.path.isdir(pretrained_model_name_or_path) and os.path.exists(
                os.path.join(pretrained_model_name_or_path, "mp_rank_00")):
            # TP-sharded checkpoint: load this rank's shard (same TP size) or
            # merge all shards into a full model (TP off)
            from ..parallel import topo

            with open(os.path.join(pretrained_model_name_or_path, "config.json")) as f:
                hf_cfg = json.load(f)
            from .nn.convert import config_from_hf

            cfg = config_from_hf(hf_cfg)
            saved_tp = int(hf_cfg.get("trlx_amd_tp_size", 1))
            base = CausalTransformer(cfg)
            model = cls(base, **wrapped_kwargs)
            if topo.tp_size() == saved_tp and saved_tp > 1:
                shard = torch.load(
                    os.path.join(pretrained_model_name_or_path,
                                 f"mp_rank_{topo.tp_rank():02d}", "model_weights.pt"),
                    map_location="cpu", weights_only=True)
                model.load_state_dict(shard, strict=False)
            else:
                # cross-TP-size resharding: merge the saved shards to a full
                # state dict, then (for tp > 1) cut this rank's new shard —
                # the NeMo-path checkpoint resharding generalized to any
                # saved/live TP pair
                from ..parallel.tp import merge_state_dicts_tp, shard_state_dict_tp

                shards = [
                    torch.load(os.path.join(pretrained_model_name_or_path,
                                            f"mp_rank_{r:02d}", "model_weights.pt"),
                               map_location="cpu", weights_only=True)
                    for r in range(saved_tp)
                ]
                full = merge_state_dicts_tp(shards, cfg, saved_tp)
                if topo.tp_size() > 1:
                    full = shard_state_dict_tp(full, cfg, topo.tp_rank(), topo.tp_size())
                model.load_state_dict(full, strict=False)
            return model
        elif os.path.isdir(pretrained_model_name_or_path):
            cfg, sd = load_hf_dir(pretrained_model_name_or_path)
            # wrapper head weights (if this directory was saved by a wrapper)
            heads_path = os.path.join(pretrained_model_name_or_path, WRAPPER_HEADS_NAME)
            if os.path.exists(heads_path):
                heads_sd = torch.load(heads_path, map_location="cpu", weights_only=True)
            base = CausalTransformer(cfg)
            missing, unexpected = base.load_state_dict(sd, strict=False)
            missing = [m for m in missing if not m.startswith("rope_")]
            if missing:
                logger.warning(f"missing keys loading {pretrained_model_name_or_path}: {missing}")
            adapter_path = os.path.join(pretrained_model_name_or_path, "adapter_model.pt")
            if os.path.exists(adapter_path) and wrapped_kwargs.get("peft_config") is not None:
                from .lora import apply_peft, load_adapter_state_dict

                apply_peft(base, wrapped_kwargs["peft_config"])
                load_adapter_state_dict(base, torch.load(adapter_path, map_location="cpu",
                                                         weights_only=True))
                wrapped_kwargs = dict(wrapped_kwargs)
                wrapped_kwargs["_lora_applied"] = True
        elif pretrained_model_name_or_path in PRESETS:
            logger.warning(
                f"'{pretrained_model_name_or_path}' is not a local directory; building a "
                f"randomly-initialized '{pretrained_model_name_or_path}' preset (no network access)."
            )
            base = CausalTransformer(preset(pretrained_model_name_or_path))
        else:
            raise OSError(
                f"'{pretrained_model_name_or_path}' is neither a local HF directory nor a known "
                f"preset ({sorted(PRESETS)}); there is no network access to fetch it."
            )
        if wrapped_kwargs.pop("_lora_applied", False):
            pass
        elif wrapped_kwargs.get("peft_config") is not None:
            from .lora import apply_peft

            apply_peft(base, wrapped_kwargs["peft_config"])
        model = cls(base, **wrapped_kwargs)
        if heads_sd:
            model.post_init(heads_sd)
        else:
            model.post_init({})
        return model

    def post_init(self, state_dict: Dict[str, torch.Tensor]):
        """Load head weights saved by ``save_pretrained`` (no-op otherwise)."""
        if not state_dict:
            return
        own = {}
        for k, v in state_dict.items():
            own[k] = v
        missing, unexpected = self.load_state_dict(own, strict=False)
        if unexpected:
            logger.warning(f"unexpected head keys: {unexpected}")

    # --- persistence --------------------------------------------------------

    def heads_state_dict(self) -> Dict[str, torch.Tensor]:
        """State dict of everything except the base model (prefixed keys)."""
        full = self.state_dict()
        return {k: v for k, v in full.items() if not k.startswith("base_model.")}

    def save_pretrained(self, save_directory: str, **kwargs):
        """Write an HF directory for the base model + wrapper head weights.
        Under LoRA, the adapter is saved separately (adapter_model.pt, the
        reference's peft behavior — modeling_base.py:328-355) and the exported
        base weights exclude the adapter.  Under tensor parallelism each TP
        rank writes its shard to mp_rank_XX/ (the NeMo format,
        modeling_nemo_ppo.py:445-467); from_pretrained reassembles."""
        from .lora import adapter_state_dict, has_adapter
        from ..parallel import topo

        if topo.tp_size() > 1:
            os.makedirs(save_directory, exist_ok=True)
            shard_dir = os.path.join(save_directory, f"mp_rank_{topo.tp_rank():02d}")
            os.makedirs(shard_dir, exist_ok=True)
            torch.save({k: v.cpu() for k, v in self.state_dict().items()
                        if not ("rope_cos" in k or "rope_sin" in k)},
                       os.path.join(shard_dir, "model_weights.pt"))
            if topo.tp_rank() == 0:
                with open(os.path.join(save_directory, "config.json"), "w") as f:
                    json.dump({**config_to_hf(self.config),
                               "trlx_amd_tp_size": topo.tp_size()}, f, indent=2)
            return

        os.makedirs(save_directory, exist_ok=True)
        base_sd = {}
        for k, v in self.base_model.state_dict().items():
            if "lora_A" in k or "lora_B" in k:
                continue
            base_sd[k] = v.cpu()
        # LoRALinear nests the original module as `.base`
        base_sd = {k.replace(".base.weight", ".weight").replace(".base.bias", ".bias"): v
                   for k, v in base_sd.items()}
        save_hf_dir(save_directory, self.config, base_sd)
        if has_adapter(self.base_model):
            torch.save(adapter_state_dict(self.base_model),
                       os.path.join(save_directory, "adapter_model.pt"))
        heads = {k: v.cpu() for k, v in self.heads_state_dict().items()}
        if heads:
            torch.save(heads, os.path.join(save_directory, WRAPPER_HEADS_NAME))

    # --- misc ---------------------------------------------------------------

    def cast_compute(self, dtype):
        """Cast the whole compute path (trunk, frozen branch, heads) to
        ``dtype``; RoPE tables stay fp32 (the kernels require it).  fp32
        master weights live in the optimizer (built after this cast).
        Matches the reference, where heads follow the model dtype
        (modeling_ilql.py:286-289)."""
        self.to(dtype)
        if self.base_model.rope_tables is not None:
            self.base_model.rope_cos = self.base_model.rope_cos.float()
            self.base_model.rope_sin = self.base_model.rope_sin.float()
        return self

    @property
    def device(self):
        return next(self.parameters()).device

    @property
    def dtype(self):
        return next(self.base_model.parameters()).dtype

    def num_parameters(self) -> int:
        return sum(p.numel() for p in self.parameters())
