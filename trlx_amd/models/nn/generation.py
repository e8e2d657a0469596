"""Autoregressive generation on the native transformer.

Replaces HF ``model.generate`` in the rollout hot path:
preallocated contiguous KV cache, one fused flash-decode attention kernel per
layer per token, and fused Gumbel-max sampling — no per-step softmax / filter
tensor materialization.  Left-padded prompts are handled with per-row position
ids and key-start offsets.

``shaping_fn(logits, hidden, last_tokens) -> logits`` hooks ILQL's
``pi + beta*(minQ - V)`` logit shaping (reference modeling_ilql.py:325-412)
and task logit masks (randomwalks) into the same loop.
"""

from dataclasses import dataclass, field
from typing import Callable, Optional

import torch

from ... import ops


@dataclass
class GenerateConfig:
    max_new_tokens: int = 40
    min_new_tokens: int = 0
    do_sample: bool = True
    temperature: float = 1.0
    top_k: int = 0
    top_p: float = 1.0
    eos_token_id: Optional[int] = None
    pad_token_id: Optional[int] = None
    seed: Optional[int] = None

    @classmethod
    def from_kwargs(cls, **kwargs) -> "GenerateConfig":
        kwargs = dict(kwargs)
        kwargs.pop("max_length", None)
        known = {f for f in cls.__dataclass_fields__}
        return cls(**{k: v for k, v in kwargs.items() if k in known})


@torch.no_grad()
def generate(
    model,
    input_ids: torch.Tensor,
    attention_mask: Optional[torch.Tensor] = None,
    gen: Optional[GenerateConfig] = None,
    shaping_fn: Optional[Callable] = None,
    **kwargs,
) -> torch.Tensor:
    """Returns [B, T_prompt + T_gen] (prompt included, HF convention).

    Rows that emit EOS are frozen and padded with ``pad_token_id``.
    """
    if gen is None:
        gen = GenerateConfig.from_kwargs(**kwargs)
    B, T = input_ids.shape
    device = input_ids.device
    was_training = model.training
    model.eval()

    if attention_mask is None:
        attention_mask = torch.ones_like(input_ids)
    key_starts = (T - attention_mask.sum(-1)).to(torch.int32)

    kv = model.new_kv_cache(B, T + gen.max_new_tokens, device=device)

    out = model(
        input_ids,
        attention_mask=attention_mask,
        kv_cache=kv,
        start_pos=0,
        return_logits=False,
    )
    # only the last position's logits are needed: lm_head on [B, 1, H]
    logits_last = model.lm_head(out.last_hidden_state[:, -1:, :])[:, 0]
    hidden_last = out.last_hidden_state[:, -1]

    pad_id = gen.pad_token_id
    if pad_id is None:
        pad_id = gen.eos_token_id if gen.eos_token_id is not None else 0
    eos_id = gen.eos_token_id

    seed = gen.seed
    if seed is None:
        seed = int(torch.randint(0, 2**31 - 1, (1,)).item())

    finished = torch.zeros(B, dtype=torch.bool, device=device)
    generated = []
    last_tokens = input_ids[:, -1]
    for step in range(gen.max_new_tokens):
        logits = logits_last.float()
        if shaping_fn is not None:
            logits = shaping_fn(logits, hidden_last, last_tokens)
        if eos_id is not None and step < gen.min_new_tokens:
            logits[:, eos_id] = float("-inf")
        if gen.do_sample:
            next_tok = ops.sample_token(
                logits, gen.temperature, gen.top_k, gen.top_p, seed=seed, offset=step
            )
        else:
            next_tok = logits.argmax(dim=-1)
        next_tok = torch.where(finished, torch.full_like(next_tok, pad_id), next_tok)
        generated.append(next_tok)
        if eos_id is not None:
            finished = finished | (next_tok == eos_id)
            if bool(finished.all()):
                break
        last_tokens = next_tok
        if step == gen.max_new_tokens - 1:
            break
        start_pos = T + step
        position_ids = (start_pos - key_starts).to(torch.int32).unsqueeze(1)
        seq_lens = torch.full((B,), start_pos + 1, dtype=torch.int32, device=device)
        out = model(
            next_tok.unsqueeze(1),
            kv_cache=kv,
            start_pos=start_pos,
            position_ids=position_ids,
            seq_lens=seq_lens,
            key_starts=key_starts,
            return_logits=False,
        )
        logits_last = model.lm_head(out.last_hidden_state[:, -1:, :])[:, 0]
        hidden_last = out.last_hidden_state[:, -1]

    if was_training:
        model.train()
    if not generated:
        return input_ids
    return torch.cat([input_ids, torch.stack(generated, dim=1)], dim=1)
