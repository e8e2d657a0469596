"""Native T5-family encoder-decoder transformer.

Parity target: the reference's seq2seq support (SURVEY.md C11 seq2seq
wrappers, flan-t5 examples).  Architecture follows T5: RMSNorm (the fused HIP
kernel), relative-position-bucket attention bias held by layer 0 of each
stack, UNSCALED q@k^T scores, relu or gated-gelu FFN, and the d_model^-0.5
output rescale under tied embeddings.

Attention here runs eager (rocBLAS GEMMs + additive-bias softmax) because the
T5 bias breaks the causal-softmax kernel's mask-free contract; norms, logprob
gathers and sampling still hit the gfx950 kernels.  Decoder generation keeps a
KV cache and precomputes cross-attention K/V once per prompt.
"""

import math
from dataclasses import dataclass, field
from typing import Any, Dict, Optional, Tuple

import torch
import torch.nn as nn
import torch.nn.functional as F

from ... import ops


@dataclass
class Seq2SeqConfig:
    vocab_size: int = 32128
    d_model: int = 512
    d_kv: int = 64
    num_heads: int = 8
    d_ff: int = 2048
    num_layers: int = 6
    num_decoder_layers: Optional[int] = None
    relative_attention_num_buckets: int = 32
    relative_attention_max_distance: int = 128
    layer_norm_epsilon: float = 1e-6
    feed_forward_proj: str = "relu"  # "relu" | "gated-gelu"
    dropout_rate: float = 0.0
    tie_word_embeddings: bool = True
    decoder_start_token_id: int = 0
    pad_token_id: int = 0
    eos_token_id: int = 1
    arch_name: str = "t5"
    extra: Dict[str, Any] = field(default_factory=dict)

    def __post_init__(self):
        if self.num_decoder_layers is None:
            self.num_decoder_layers = self.num_layers

    @property
    def is_gated(self) -> bool:
        return self.feed_forward_proj.startswith("gated")

    def to_dict(self):
        return dict(self.__dict__)

    @classmethod
    def from_dict(cls, d):
        return cls(**d)


def relative_position_bucket(relative_position: torch.Tensor, bidirectional: bool,
                             num_buckets: int, max_distance: int) -> torch.Tensor:
    """T5's log-bucketed relative positions (semantics of the T5 paper)."""
    ret = torch.zeros_like(relative_position)
    n = -relative_position
    if bidirectional:
        num_buckets //= 2
        ret = ret + (n < 0).long() * num_buckets
        n = n.abs()
    else:
        n = torch.clamp(n, min=0)
    max_exact = num_buckets // 2
    is_small = n < max_exact
    val_large = max_exact + (
        torch.log(n.float() / max_exact + 1e-9) / math.log(max_distance / max_exact)
        * (num_buckets - max_exact)
    ).long()
    val_large = torch.clamp(val_large, max=num_buckets - 1)
    return ret + torch.where(is_small, n, val_large)


class T5Norm(nn.Module):
    """T5LayerNorm == RMSNorm (the fused HIP kernel path)."""

    def __init__(self, d_model: int, eps: float):
        super().__init__()
        self.weight = nn.Parameter(torch.ones(d_model))
        self.eps = eps

    def forward(self, x):
        return ops.rmsnorm(x, self.weight, self.eps)


class T5Attention(nn.Module):
    def __init__(self, cfg: Seq2SeqConfig, has_bias: bool, bidirectional: bool):
        super().__init__()
        inner = cfg.num_heads * cfg.d_kv
        self.q = nn.Linear(cfg.d_model, inner, bias=False)
        self.k = nn.Linear(cfg.d_model, inner, bias=False)
        self.v = nn.Linear(cfg.d_model, inner, bias=False)
        self.o = nn.Linear(inner, cfg.d_model, bias=False)
        self.num_heads = cfg.num_heads
        self.d_kv = cfg.d_kv
        self.bidirectional = bidirectional
        self.cfg = cfg
        self.relative_attention_bias = (
            nn.Embedding(cfg.relative_attention_num_buckets, cfg.num_heads) if has_bias else None
        )

    def compute_bias(self, q_len: int, k_len: int, device, q_offset: int = 0) -> torch.Tensor:
        """[1, H, Tq, Tk] additive bias."""
        ctx = torch.arange(q_len, device=device)[:, None] + q_offset
        mem = torch.arange(k_len, device=device)[None, :]
        rel = mem - ctx
        buckets = relative_position_bucket(
            rel, self.bidirectional, self.cfg.relative_attention_num_buckets,
            self.cfg.relative_attention_max_distance,
        )
        bias = self.relative_attention_bias(buckets)  # [Tq, Tk, H]
        return bias.permute(2, 0, 1).unsqueeze(0)

    def _shape(self, x, B, T):
        return x.view(B, T, self.num_heads, self.d_kv).transpose(1, 2)

    def forward(self, x, kv_input=None, mask=None, position_bias=None, past_kv=None):
        """``mask``: additive float mask [B, 1, Tq, Tk] (0 / -inf).  Returns
        (out, position_bias, new_kv)."""
        B, Tq, _ = x.shape
        q = self._shape(self.q(x), B, Tq)
        if kv_input is None:  # self-attention
            k = self._shape(self.k(x), B, Tq)
            v = self._shape(self.v(x), B, Tq)
            if past_kv is not None:
                k = torch.cat([past_kv[0], k], dim=2)
                v = torch.cat([past_kv[1], v], dim=2)
            new_kv = (k, v)
        elif past_kv is not None:  # cached cross-attention
            k, v = past_kv
            new_kv = past_kv
        else:
            Tk = kv_input.shape[1]
            k = self._shape(self.k(kv_input), B, Tk)
            v = self._shape(self.v(kv_input), B, Tk)
            new_kv = (k, v)

        # T5: NO 1/sqrt(d) scaling
        scores = torch.matmul(q, k.transpose(-1, -2)).float()
        if position_bias is None and self.relative_attention_bias is not None:
            q_offset = k.shape[2] - Tq
            position_bias = self.compute_bias(Tq, k.shape[2], x.device, q_offset=q_offset)
        if position_bias is not None:
            scores = scores + position_bias.float()
        if mask is not None:
            scores = scores + mask.float()
        probs = torch.softmax(scores, dim=-1).to(v.dtype)
        out = torch.matmul(probs, v).transpose(1, 2).reshape(B, Tq, -1)
        return self.o(out), position_bias, new_kv


class T5FFN(nn.Module):
    def __init__(self, cfg: Seq2SeqConfig):
        super().__init__()
        if cfg.is_gated:
            self.wi_0 = nn.Linear(cfg.d_model, cfg.d_ff, bias=False)
            self.wi_1 = nn.Linear(cfg.d_model, cfg.d_ff, bias=False)
        else:
            self.wi = nn.Linear(cfg.d_model, cfg.d_ff, bias=False)
        self.wo = nn.Linear(cfg.d_ff, cfg.d_model, bias=False)
        self.gated = cfg.is_gated

    def forward(self, x):
        if self.gated:
            h = F.gelu(self.wi_0(x), approximate="tanh") * self.wi_1(x)
        else:
            h = F.relu(self.wi(x))
        return self.wo(h)


class T5Block(nn.Module):
    def __init__(self, cfg: Seq2SeqConfig, is_decoder: bool, has_bias: bool):
        super().__init__()
        self.is_decoder = is_decoder
        self.self_norm = T5Norm(cfg.d_model, cfg.layer_norm_epsilon)
        self.self_attn = T5Attention(cfg, has_bias, bidirectional=not is_decoder)
        if is_decoder:
            self.cross_norm = T5Norm(cfg.d_model, cfg.layer_norm_epsilon)
            self.cross_attn = T5Attention(cfg, has_bias=False, bidirectional=True)
        self.ffn_norm = T5Norm(cfg.d_model, cfg.layer_norm_epsilon)
        self.ffn = T5FFN(cfg)

    def forward(self, x, enc_out=None, self_mask=None, cross_mask=None, position_bias=None,
                cross_bias=None, past_self_kv=None, past_cross_kv=None):
        h, position_bias, new_self_kv = self.self_attn(
            self.self_norm(x), mask=self_mask, position_bias=position_bias, past_kv=past_self_kv)
        x = x + h
        new_cross_kv = None
        if self.is_decoder and enc_out is not None:
            h, _, new_cross_kv = self.cross_attn(
                self.cross_norm(x), kv_input=enc_out, mask=cross_mask, position_bias=cross_bias,
                past_kv=past_cross_kv)
            x = x + h
        x = x + self.ffn(self.ffn_norm(x))
        return x, position_bias, new_self_kv, new_cross_kv


@dataclass
class Seq2SeqOutput:
    logits: Optional[torch.Tensor] = None
    last_hidden_state: Optional[torch.Tensor] = None
    encoder_last_hidden_state: Optional[torch.Tensor] = None
    decoder_hidden_at_layer: Optional[torch.Tensor] = None


def _extend_mask(mask: Optional[torch.Tensor], dtype=torch.float32) -> Optional[torch.Tensor]:
    """[B, T] 1/0 -> additive [B, 1, 1, T]."""
    if mask is None:
        return None
    return (1.0 - mask[:, None, None, :].float()) * torch.finfo(dtype).min


class Seq2SeqTransformer(nn.Module):
    """T5-family encoder-decoder LM."""

    def __init__(self, config: Seq2SeqConfig):
        super().__init__()
        self.config = config
        cfg = config
        self.shared = nn.Embedding(cfg.vocab_size, cfg.d_model)
        self.encoder_blocks = nn.ModuleList(
            T5Block(cfg, is_decoder=False, has_bias=(i == 0)) for i in range(cfg.num_layers)
        )
        self.encoder_final_norm = T5Norm(cfg.d_model, cfg.layer_norm_epsilon)
        self.decoder_blocks = nn.ModuleList(
            T5Block(cfg, is_decoder=True, has_bias=(i == 0)) for i in range(cfg.num_decoder_layers)
        )
        self.decoder_final_norm = T5Norm(cfg.d_model, cfg.layer_norm_epsilon)
        self.lm_head = nn.Linear(cfg.d_model, cfg.vocab_size, bias=False)
        if cfg.tie_word_embeddings:
            self.lm_head.weight = self.shared.weight
        self.apply(self._init_weights)

    def _init_weights(self, module):
        if isinstance(module, nn.Linear):
            module.weight.data.normal_(0.0, 0.02)
        elif isinstance(module, nn.Embedding):
            module.weight.data.normal_(0.0, 0.02)

    # --- encoder ------------------------------------------------------------

    def encode(self, input_ids, attention_mask=None):
        h = self.shared(input_ids)
        mask = _extend_mask(attention_mask)
        bias = None
        for block in self.encoder_blocks:
            h, bias, _, _ = block(h, self_mask=mask, position_bias=bias)
        return self.encoder_final_norm(h)

    # --- decoder ------------------------------------------------------------

    def decode(self, decoder_input_ids, enc_out, attention_mask=None,
               decoder_attention_mask=None, past=None, hidden_at_layer=None):
        """``past``: list of (self_kv, cross_kv) per layer for incremental
        decoding.  Returns (hidden, new_past, hidden_at)."""
        B, T = decoder_input_ids.shape
        h = self.shared(decoder_input_ids)
        past_len = past[0][0][0].shape[2] if past is not None and past[0][0] is not None else 0
        # causal mask over [T, past+T]
        total = past_len + T
        causal = torch.ones(T, total, device=h.device).tril(diagonal=past_len)
        self_mask = (1.0 - causal[None, None]) * torch.finfo(torch.float32).min
        if decoder_attention_mask is not None:
            self_mask = self_mask + _extend_mask(decoder_attention_mask)
        cross_mask = _extend_mask(attention_mask)

        bias = None
        new_past = []
        n = len(self.decoder_blocks)
        # int -> single stash (tensor); list -> dict keyed by requested values
        multi = isinstance(hidden_at_layer, (list, tuple))
        wanted = list(hidden_at_layer) if multi else (
            [hidden_at_layer] if hidden_at_layer is not None else [])
        stash_for = {}
        for w in wanted:
            stash_for.setdefault(w % n, []).append(w)
        hidden_at = {} if multi else None
        for i, block in enumerate(self.decoder_blocks):
            if i in stash_for:
                if multi:
                    for w in stash_for[i]:
                        hidden_at[w] = h
                else:
                    hidden_at = h
            p_self = past[i][0] if past is not None else None
            p_cross = past[i][1] if past is not None else None
            h, bias, self_kv, cross_kv = block(
                h, enc_out=enc_out, self_mask=self_mask, cross_mask=cross_mask,
                position_bias=bias, past_self_kv=p_self, past_cross_kv=p_cross)
            new_past.append((self_kv, cross_kv))
        return self.decoder_final_norm(h), new_past, hidden_at

    def project(self, hidden):
        if self.config.tie_word_embeddings:
            hidden = hidden * (self.config.d_model ** -0.5)
        return self.lm_head(hidden)

    # --- full forward ---------------------------------------------------------

    def forward(self, input_ids, attention_mask=None, decoder_input_ids=None,
                decoder_attention_mask=None, hidden_at_layer=None, return_logits=True,
                logits_slice=None):
        enc = self.encode(input_ids, attention_mask)
        dec, _, hidden_at = self.decode(decoder_input_ids, enc, attention_mask,
                                        decoder_attention_mask, hidden_at_layer=hidden_at_layer)
        logits = None
        if return_logits:
            ds = dec if logits_slice is None else dec[:, logits_slice[0] : logits_slice[1]]
            logits = self.project(ds)
        return Seq2SeqOutput(logits=logits, last_hidden_state=dec,
                             encoder_last_hidden_state=enc, decoder_hidden_at_layer=hidden_at)

    # --- generation -------------------------------------------------------------

    @torch.no_grad()
    def generate(self, input_ids, attention_mask=None, max_new_tokens: int = 32,
                 do_sample: bool = True, temperature: float = 1.0, top_k: int = 0,
                 top_p: float = 1.0, eos_token_id: Optional[int] = None,
                 pad_token_id: Optional[int] = None, seed: Optional[int] = None,
                 shaping_fn=None, **kwargs):
        """Returns decoder token ids [B, 1 + T_gen] starting with
        decoder_start_token_id (HF seq2seq convention)."""
        cfg = self.config
        B = input_ids.shape[0]
        device = input_ids.device
        eos = eos_token_id if eos_token_id is not None else cfg.eos_token_id
        pad = pad_token_id if pad_token_id is not None else cfg.pad_token_id
        if seed is None:
            seed = int(torch.randint(0, 2**31 - 1, (1,)).item())

        enc = self.encode(input_ids, attention_mask)
        cur = torch.full((B, 1), cfg.decoder_start_token_id, dtype=torch.long, device=device)
        out_tokens = [cur]
        finished = torch.zeros(B, dtype=torch.bool, device=device)
        past = None
        for step in range(max_new_tokens):
            h, past, _ = self.decode(cur, enc, attention_mask, past=past)
            logits = self.project(h[:, -1:, :])[:, 0].float()
            if shaping_fn is not None:
                logits = shaping_fn(logits, h[:, -1], cur[:, -1])
            if do_sample:
                tok = ops.sample_token(logits, temperature, top_k, top_p, seed=seed, offset=step)
            else:
                tok = logits.argmax(-1)
            tok = torch.where(finished, torch.full_like(tok, pad), tok)
            out_tokens.append(tok.unsqueeze(1))
            finished = finished | (tok == eos)
            cur = tok.unsqueeze(1)
            if bool(finished.all()):
                break
        return torch.cat(out_tokens, dim=1)

    def num_parameters(self):
        return sum(p.numel() for p in self.parameters())


# ---------------------------------------------------------------------------
# HF interop
# ---------------------------------------------------------------------------


def seq2seq_config_from_hf(hf: dict) -> Seq2SeqConfig:
    proj = hf.get("feed_forward_proj", "relu")
    return Seq2SeqConfig(
        vocab_size=hf["vocab_size"],
        d_model=hf["d_model"],
        d_kv=hf["d_kv"],
        num_heads=hf["num_heads"],
        d_ff=hf["d_ff"],
        num_layers=hf["num_layers"],
        num_decoder_layers=hf.get("num_decoder_layers", hf["num_layers"]),
        relative_attention_num_buckets=hf.get("relative_attention_num_buckets", 32),
        relative_attention_max_distance=hf.get("relative_attention_max_distance", 128),
        layer_norm_epsilon=hf.get("layer_norm_epsilon", 1e-6),
        feed_forward_proj=proj,
        tie_word_embeddings=hf.get("tie_word_embeddings", True),
        decoder_start_token_id=hf.get("decoder_start_token_id", 0),
        pad_token_id=hf.get("pad_token_id", 0),
        eos_token_id=hf.get("eos_token_id", 1),
    )


def seq2seq_config_to_hf(cfg: Seq2SeqConfig) -> dict:
    return {
        "model_type": "t5",
        "architectures": ["T5ForConditionalGeneration"],
        "vocab_size": cfg.vocab_size,
        "d_model": cfg.d_model,
        "d_kv": cfg.d_kv,
        "num_heads": cfg.num_heads,
        "d_ff": cfg.d_ff,
        "num_layers": cfg.num_layers,
        "num_decoder_layers": cfg.num_decoder_layers,
        "relative_attention_num_buckets": cfg.relative_attention_num_buckets,
        "relative_attention_max_distance": cfg.relative_attention_max_distance,
        "layer_norm_epsilon": cfg.layer_norm_epsilon,
        "feed_forward_proj": cfg.feed_forward_proj,
        "tie_word_embeddings": cfg.tie_word_embeddings,
        "decoder_start_token_id": cfg.decoder_start_token_id,
        "pad_token_id": cfg.pad_token_id,
        "eos_token_id": cfg.eos_token_id,
    }


def seq2seq_state_dict_from_hf(cfg: Seq2SeqConfig, hf: Dict[str, torch.Tensor]) -> Dict[str, torch.Tensor]:
    out = {"shared.weight": hf["shared.weight"]}

    def attn(dst, src):
        for a, b in (("q", "q"), ("k", "k"), ("v", "v"), ("o", "o")):
            out[f"{dst}.{a}.weight"] = hf[f"{src}.{a}.weight"]

    for i in range(cfg.num_layers):
        src = f"encoder.block.{i}.layer"
        dst = f"encoder_blocks.{i}"
        attn(f"{dst}.self_attn", f"{src}.0.SelfAttention")
        if i == 0:
            out[f"{dst}.self_attn.relative_attention_bias.weight"] = hf[
                f"{src}.0.SelfAttention.relative_attention_bias.weight"]
        out[f"{dst}.self_norm.weight"] = hf[f"{src}.0.layer_norm.weight"]
        if cfg.is_gated:
            out[f"{dst}.ffn.wi_0.weight"] = hf[f"{src}.1.DenseReluDense.wi_0.weight"]
            out[f"{dst}.ffn.wi_1.weight"] = hf[f"{src}.1.DenseReluDense.wi_1.weight"]
        else:
            out[f"{dst}.ffn.wi.weight"] = hf[f"{src}.1.DenseReluDense.wi.weight"]
        out[f"{dst}.ffn.wo.weight"] = hf[f"{src}.1.DenseReluDense.wo.weight"]
        out[f"{dst}.ffn_norm.weight"] = hf[f"{src}.1.layer_norm.weight"]
    for i in range(cfg.num_decoder_layers):
        src = f"decoder.block.{i}.layer"
        dst = f"decoder_blocks.{i}"
        attn(f"{dst}.self_attn", f"{src}.0.SelfAttention")
        if i == 0:
            out[f"{dst}.self_attn.relative_attention_bias.weight"] = hf[
                f"{src}.0.SelfAttention.relative_attention_bias.weight"]
        out[f"{dst}.self_norm.weight"] = hf[f"{src}.0.layer_norm.weight"]
        attn(f"{dst}.cross_attn", f"{src}.1.EncDecAttention")
        out[f"{dst}.cross_norm.weight"] = hf[f"{src}.1.layer_norm.weight"]
        if cfg.is_gated:
            out[f"{dst}.ffn.wi_0.weight"] = hf[f"{src}.2.DenseReluDense.wi_0.weight"]
            out[f"{dst}.ffn.wi_1.weight"] = hf[f"{src}.2.DenseReluDense.wi_1.weight"]
        else:
            out[f"{dst}.ffn.wi.weight"] = hf[f"{src}.2.DenseReluDense.wi.weight"]
        out[f"{dst}.ffn.wo.weight"] = hf[f"{src}.2.DenseReluDense.wo.weight"]
        out[f"{dst}.ffn_norm.weight"] = hf[f"{src}.2.layer_norm.weight"]
    out["encoder_final_norm.weight"] = hf["encoder.final_layer_norm.weight"]
    out["decoder_final_norm.weight"] = hf["decoder.final_layer_norm.weight"]
    if not cfg.tie_word_embeddings and "lm_head.weight" in hf:
        out["lm_head.weight"] = hf["lm_head.weight"]
    return out


def seq2seq_state_dict_to_hf(cfg: Seq2SeqConfig, sd: Dict[str, torch.Tensor]) -> Dict[str, torch.Tensor]:
    out = {
        "shared.weight": sd["shared.weight"],
        "encoder.embed_tokens.weight": sd["shared.weight"],
        "decoder.embed_tokens.weight": sd["shared.weight"],
        "lm_head.weight": sd.get("lm_head.weight", sd["shared.weight"]),
        "encoder.final_layer_norm.weight": sd["encoder_final_norm.weight"],
        "decoder.final_layer_norm.weight": sd["decoder_final_norm.weight"],
    }

    def attn(dst, src):
        for a in "qkvo":
            out[f"{dst}.{a}.weight"] = sd[f"{src}.{a}.weight"]

    for i in range(cfg.num_layers):
        dst = f"encoder.block.{i}.layer"
        src = f"encoder_blocks.{i}"
        attn(f"{dst}.0.SelfAttention", f"{src}.self_attn")
        if i == 0:
            out[f"{dst}.0.SelfAttention.relative_attention_bias.weight"] = sd[
                f"{src}.self_attn.relative_attention_bias.weight"]
        out[f"{dst}.0.layer_norm.weight"] = sd[f"{src}.self_norm.weight"]
        if cfg.is_gated:
            out[f"{dst}.1.DenseReluDense.wi_0.weight"] = sd[f"{src}.ffn.wi_0.weight"]
            out[f"{dst}.1.DenseReluDense.wi_1.weight"] = sd[f"{src}.ffn.wi_1.weight"]
        else:
            out[f"{dst}.1.DenseReluDense.wi.weight"] = sd[f"{src}.ffn.wi.weight"]
        out[f"{dst}.1.DenseReluDense.wo.weight"] = sd[f"{src}.ffn.wo.weight"]
        out[f"{dst}.1.layer_norm.weight"] = sd[f"{src}.ffn_norm.weight"]
    for i in range(cfg.num_decoder_layers):
        dst = f"decoder.block.{i}.layer"
        src = f"decoder_blocks.{i}"
        attn(f"{dst}.0.SelfAttention", f"{src}.self_attn")
        if i == 0:
            out[f"{dst}.0.SelfAttention.relative_attention_bias.weight"] = sd[
                f"{src}.self_attn.relative_attention_bias.weight"]
        out[f"{dst}.0.layer_norm.weight"] = sd[f"{src}.self_norm.weight"]
        attn(f"{dst}.1.EncDecAttention", f"{src}.cross_attn")
        out[f"{dst}.1.layer_norm.weight"] = sd[f"{src}.cross_norm.weight"]
        if cfg.is_gated:
            out[f"{dst}.2.DenseReluDense.wi_0.weight"] = sd[f"{src}.ffn.wi_0.weight"]
            out[f"{dst}.2.DenseReluDense.wi_1.weight"] = sd[f"{src}.ffn.wi_1.weight"]
        else:
            out[f"{dst}.2.DenseReluDense.wi.weight"] = sd[f"{src}.ffn.wi.weight"]
        out[f"{dst}.2.DenseReluDense.wo.weight"] = sd[f"{src}.ffn.wo.weight"]
        out[f"{dst}.2.layer_norm.weight"] = sd[f"{src}.ffn_norm.weight"]
    return out
