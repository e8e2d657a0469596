"""Seq2seq (T5-family) method wrappers.

Parity target: reference trlx/models/modeling_ppo.py seq2seq classes
(AutoModelForSeq2SeqLMWithValueHead 1242-1350, ...HydraValueHead 1353-1480,
T5Branch 1483-1592) and modeling_ilql.py seq2seq ILQL heads (445-666), on the
native Seq2SeqTransformer.
"""

import copy
import json
import os
from typing import Optional

import torch
import torch.nn as nn

from ..utils import logging
from ..utils.modeling import make_head
from .modeling_ilql import ILQLHeads
from .modeling_ppo import CausalLMOutputWithValue
from .nn.seq2seq import (
    Seq2SeqConfig,
    Seq2SeqTransformer,
    seq2seq_config_from_hf,
    seq2seq_config_to_hf,
    seq2seq_state_dict_from_hf,
    seq2seq_state_dict_to_hf,
)

logger = logging.get_logger(__name__)

WRAPPER_HEADS_NAME = "wrapper_heads.pt"


class Seq2SeqModelWrapper(nn.Module):
    """from_pretrained / from_config / save_pretrained for seq2seq wrappers
    (same checkpoint contract as the causal PreTrainedModelWrapper)."""

    def __init__(self, base_model: Seq2SeqTransformer, **kwargs):
        super().__init__()
        self.base_model = base_model
        self.config = base_model.config

    @classmethod
    def from_config(cls, config, **kwargs):
        if isinstance(config, dict):
            config = Seq2SeqConfig.from_dict(config)
        return cls(Seq2SeqTransformer(config), **kwargs)

    @classmethod
    def from_pretrained(cls, path, **kwargs):
        if isinstance(path, Seq2SeqTransformer):
            return cls(path, **kwargs)
        if not os.path.isdir(path):
            raise OSError(f"'{path}' is not a local HF directory (no network access)")
        with open(os.path.join(path, "config.json")) as f:
            hf_cfg = json.load(f)
        cfg = seq2seq_config_from_hf(hf_cfg)
        import safetensors.torch

        st = os.path.join(path, "model.safetensors")
        pt = os.path.join(path, "pytorch_model.bin")
        if os.path.exists(st):
            sd = safetensors.torch.load_file(st)
        else:
            sd = torch.load(pt, map_location="cpu", weights_only=True)
        base = Seq2SeqTransformer(cfg)
        base.load_state_dict(seq2seq_state_dict_from_hf(cfg, sd), strict=False)
        model = cls(base, **kwargs)
        heads_path = os.path.join(path, WRAPPER_HEADS_NAME)
        if os.path.exists(heads_path):
            model.load_state_dict(torch.load(heads_path, map_location="cpu", weights_only=True),
                                  strict=False)
        return model

    def save_pretrained(self, directory: str, **kwargs):
        os.makedirs(directory, exist_ok=True)
        with open(os.path.join(directory, "config.json"), "w") as f:
            json.dump(seq2seq_config_to_hf(self.config), f, indent=2)
        sd = {k: v.cpu() for k, v in self.base_model.state_dict().items()}
        hf_sd = seq2seq_state_dict_to_hf(self.config, sd)
        seen = {}
        for k, v in list(hf_sd.items()):
            if v.data_ptr() in seen:
                hf_sd[k] = v.clone()
            seen[v.data_ptr()] = k
        import safetensors.torch

        safetensors.torch.save_file({k: v.contiguous() for k, v in hf_sd.items()},
                                    os.path.join(directory, "model.safetensors"))
        heads = {k: v.cpu() for k, v in self.state_dict().items()
                 if not k.startswith("base_model.")}
        if heads:
            torch.save(heads, os.path.join(directory, WRAPPER_HEADS_NAME))

    def cast_compute(self, dtype):
        self.to(dtype)
        return self

    def generate(self, input_ids, attention_mask=None, **kwargs):
        return self.base_model.generate(input_ids, attention_mask, **kwargs)

    def generate_eval(self, input_ids, attention_mask=None, **kwargs):
        return self.generate(input_ids, attention_mask, **kwargs)


class T5ValueBranch(nn.Module):
    """TRAINABLE value branch for seq2seq: copies of the top
    ``num_value_layers_unfrozen`` decoder blocks + final norm + scalar head
    (reference make_value_branch applied to the T5 branch class,
    modeling_ppo.py:255-263/1483-1592)."""

    def __init__(self, base: Seq2SeqTransformer, num_layers: int):
        super().__init__()
        self.cfg = base.config
        self.num_layers = num_layers
        self.blocks = nn.ModuleList(
            copy.deepcopy(b) for b in base.decoder_blocks[-num_layers:]
        )
        self.final_norm = copy.deepcopy(base.decoder_final_norm)
        self.v_head = make_head(self.cfg.d_model, 1, dtype=torch.float32)
        for p in self.parameters():
            p.requires_grad_(True)

    def forward(self, hidden, enc_out, attention_mask, decoder_attention_mask,
                position_bias, logits_slice=None):
        from .nn.seq2seq import _extend_mask

        B, T = hidden.shape[:2]
        causal = torch.ones(T, T, device=hidden.device).tril()
        self_mask = (1.0 - causal[None, None]) * torch.finfo(torch.float32).min
        if decoder_attention_mask is not None:
            self_mask = self_mask + _extend_mask(decoder_attention_mask)
        cross_mask = _extend_mask(attention_mask)
        h = hidden
        for block in self.blocks:
            h, position_bias, _, _ = block(h, enc_out=enc_out, self_mask=self_mask,
                                           cross_mask=cross_mask, position_bias=position_bias)
        h = self.final_norm(h)
        if logits_slice is not None:
            h = h[:, logits_slice[0] : logits_slice[1]]
        return self.v_head(h.to(self.v_head[0].weight.dtype)).squeeze(-1).float()


class AutoModelForSeq2SeqLMWithValueHead(Seq2SeqModelWrapper):
    """Seq2seq LM + scalar value head on decoder hidden states
    (reference modeling_ppo.py:1242-1350)."""

    def __init__(self, base_model, peft_config=None, num_value_layers_unfrozen: int = 0):
        super().__init__(base_model)
        self.peft_config = peft_config
        self.num_value_layers_unfrozen = num_value_layers_unfrozen
        self.v_head = make_head(self.config.d_model, 1, dtype=torch.float32)
        self.v_branch = None
        if num_value_layers_unfrozen > 0:
            self.v_branch = T5ValueBranch(base_model, num_value_layers_unfrozen)

    def _value_bias(self, T, device):
        return self.base_model.decoder_blocks[0].self_attn.compute_bias(T, T, device)

    def forward(self, input_ids, attention_mask=None, decoder_input_ids=None,
                decoder_attention_mask=None, return_ref_logits: bool = False,
                logits_slice=None, **kwargs):
        if self.v_branch is not None:
            enc = self.base_model.encode(input_ids, attention_mask)
            dec, _, hidden_at = self.base_model.decode(
                decoder_input_ids, enc, attention_mask, decoder_attention_mask,
                hidden_at_layer=-self.num_value_layers_unfrozen)
            ds = dec if logits_slice is None else dec[:, logits_slice[0] : logits_slice[1]]
            logits = self.base_model.project(ds)
            values = self.v_branch(hidden_at, enc, attention_mask, decoder_attention_mask,
                                   self._value_bias(decoder_input_ids.shape[1], dec.device),
                                   logits_slice=logits_slice)
            return CausalLMOutputWithValue(logits=logits, values=values,
                                           last_hidden_state=dec)
        out = self.base_model(input_ids, attention_mask, decoder_input_ids,
                              decoder_attention_mask, logits_slice=logits_slice)
        hs = out.last_hidden_state
        if logits_slice is not None:
            hs = hs[:, logits_slice[0] : logits_slice[1]]
        values = self.v_head(hs.to(self.v_head[0].weight.dtype)).squeeze(-1).float()
        return CausalLMOutputWithValue(logits=out.logits, values=values,
                                       last_hidden_state=out.last_hidden_state)


class T5Branch(nn.Module):
    """Frozen copies of the top decoder blocks + final norm + lm_head
    (reference T5Branch, modeling_ppo.py:1483-1592)."""

    def __init__(self, base: Seq2SeqTransformer, num_layers_unfrozen: int):
        super().__init__()
        self.cfg = base.config
        self.num_layers_unfrozen = num_layers_unfrozen
        self.blocks = nn.ModuleList(
            copy.deepcopy(b) for b in base.decoder_blocks[-num_layers_unfrozen:]
        )
        self.final_norm = copy.deepcopy(base.decoder_final_norm)
        self.lm_head = copy.deepcopy(base.lm_head)
        for p in self.parameters():
            p.requires_grad_(False)

    def forward(self, hidden, enc_out, attention_mask, decoder_attention_mask,
                position_bias, logits_slice=None):
        from .nn.seq2seq import _extend_mask

        with torch.no_grad():
            B, T = hidden.shape[:2]
            causal = torch.ones(T, T, device=hidden.device).tril()
            self_mask = (1.0 - causal[None, None]) * torch.finfo(torch.float32).min
            if decoder_attention_mask is not None:
                self_mask = self_mask + _extend_mask(decoder_attention_mask)
            cross_mask = _extend_mask(attention_mask)
            h = hidden
            for block in self.blocks:
                h, position_bias, _, _ = block(h, enc_out=enc_out, self_mask=self_mask,
                                               cross_mask=cross_mask, position_bias=position_bias)
            h = self.final_norm(h)
            if logits_slice is not None:
                h = h[:, logits_slice[0] : logits_slice[1]]
            if self.cfg.tie_word_embeddings:
                h = h * (self.cfg.d_model ** -0.5)
            return self.lm_head(h)


class AutoModelForSeq2SeqLMWithHydraValueHead(AutoModelForSeq2SeqLMWithValueHead):
    """Seq2seq value head + frozen reference branch
    (reference modeling_ppo.py:1353-1480)."""

    def __init__(self, base_model, peft_config=None, num_layers_unfrozen: int = -1,
                 num_value_layers_unfrozen: int = 0):
        super().__init__(base_model, peft_config=peft_config,
                         num_value_layers_unfrozen=num_value_layers_unfrozen)
        self.num_layers_unfrozen = num_layers_unfrozen
        self.frozen_head = None
        if num_layers_unfrozen > 0 and peft_config is None:
            self.frozen_head = T5Branch(base_model, num_layers_unfrozen)

    def forward(self, input_ids, attention_mask=None, decoder_input_ids=None,
                decoder_attention_mask=None, return_ref_logits: bool = False,
                logits_slice=None, **kwargs):
        stash = []
        if return_ref_logits and self.frozen_head is not None:
            stash.append(-self.num_layers_unfrozen)
        if self.v_branch is not None:
            stash.append(-self.num_value_layers_unfrozen)
        enc = self.base_model.encode(input_ids, attention_mask)
        dec, _, hidden_at = self.base_model.decode(
            decoder_input_ids, enc, attention_mask, decoder_attention_mask,
            hidden_at_layer=stash or None)
        ds = dec if logits_slice is None else dec[:, logits_slice[0] : logits_slice[1]]
        logits = self.base_model.project(ds)
        T = decoder_input_ids.shape[1]
        if self.v_branch is not None:
            values = self.v_branch(hidden_at[-self.num_value_layers_unfrozen], enc,
                                   attention_mask, decoder_attention_mask,
                                   self._value_bias(T, dec.device), logits_slice=logits_slice)
        else:
            values = self.v_head(ds.to(self.v_head[0].weight.dtype)).squeeze(-1).float()
        ref_logits = None
        if return_ref_logits and self.frozen_head is not None:
            # recompute the first unfrozen block's incoming position bias
            bias_layer = self.base_model.decoder_blocks[0].self_attn
            position_bias = bias_layer.compute_bias(T, T, dec.device)
            ref_hidden = hidden_at[-self.num_layers_unfrozen] if isinstance(hidden_at, dict) \
                else hidden_at
            ref_logits = self.frozen_head(ref_hidden, enc, attention_mask,
                                          decoder_attention_mask, position_bias,
                                          logits_slice=logits_slice)
        return CausalLMOutputWithValue(logits=logits, values=values, ref_logits=ref_logits,
                                       last_hidden_state=dec)

    def forward_hydra(self, input_ids, attention_mask=None, decoder_input_ids=None,
                      decoder_attention_mask=None, **kwargs):
        out = self.forward(input_ids, attention_mask, decoder_input_ids,
                           decoder_attention_mask, return_ref_logits=True)
        return CausalLMOutputWithValue(logits=out.ref_logits)


class AutoModelForSeq2SeqLMWithILQLHeads(Seq2SeqModelWrapper):
    """Seq2seq ILQL heads (reference modeling_ilql.py:445-666)."""

    def __init__(self, base_model, two_qs: bool = True, alpha: float = 0.99, peft_config=None):
        super().__init__(base_model)
        self.two_qs = two_qs
        self.alpha = alpha
        self.peft_config = peft_config
        self.ilql_heads = ILQLHeads(self.config.d_model, self.config.vocab_size, two_qs, alpha)

    def forward(self, input_ids, attention_mask=None, decoder_input_ids=None,
                actions_ixs=None, states_ixs=None, **kwargs):
        from .modeling_ilql import CausalILQLOutput

        out = self.base_model(input_ids, attention_mask, decoder_input_ids)
        qs, target_qs, vs = self.ilql_heads(out.last_hidden_state, states_ixs=states_ixs,
                                            actions_ixs=actions_ixs)
        return CausalILQLOutput(out.logits, qs, target_qs, vs, out.last_hidden_state)

    def sync_target_q_heads(self):
        self.ilql_heads.sync_target_q_heads()

    @torch.no_grad()
    def generate(self, input_ids, attention_mask=None, beta: float = 1.0,
                 max_new_tokens: int = 32, max_length: int = 1024, temperature: float = 1.0,
                 top_k: int = 20, logit_mask=None, pad_token_id=None, eos_token_id=None,
                 **kwargs):
        import torch.nn.functional as F

        def shaping_fn(logits, hidden, last_tokens):
            qs, target_qs, vs = self.ilql_heads(hidden.unsqueeze(1))
            if self.two_qs:
                q = torch.minimum(target_qs[0][:, -1, :], target_qs[1][:, -1, :])
            else:
                q = target_qs[0][:, -1, :]
            v = vs[:, -1, :]
            if logit_mask is not None:
                mask = logit_mask[last_tokens.to(logit_mask.device)]
                logits = logits.masked_fill(mask.to(logits.device), float("-inf"))
            adv = (q - v).to(logits.dtype)
            return F.log_softmax(logits, -1) + beta * adv

        return self.base_model.generate(
            input_ids, attention_mask, max_new_tokens=max_new_tokens,
            do_sample=temperature > 0, temperature=temperature if temperature > 0 else 1.0,
            top_k=top_k, eos_token_id=eos_token_id, pad_token_id=pad_token_id,
            shaping_fn=shaping_fn,
        )
