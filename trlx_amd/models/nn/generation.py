"""Autoregressive generation on the native transformer.

Replaces HF ``model.generate`` in the rollout hot path.
Two GPU paths:

1. **hipGraph decode engine** (default): the whole per-token step — fused
   decode_prep (QKV split + RoPE + cache append), flash-decode attention,
   lm_head GEMM, Gumbel-max sampling, and all state advancement — is captured
   once as a hipGraph and replayed per token.  All loop state (current token,
   cache index, sequence lengths, RNG offset, output column) lives in device
   buffers, so a 12-layer GPT-2 decode step is ONE graph launch instead of
   ~150 kernel launches with host gaps (profile r01: decode was launch-bound).
   The engine (KV cache + buffers + graph) persists across generate() calls.

2. **Eager loop**: CPU, custom shaping fns (ILQL logit shaping, logit masks),
   or TRLX_AMD_NO_GRAPHS=1.

Left-padded prompts are handled with per-row key-start offsets everywhere.
"""

import os
import weakref
from dataclasses import dataclass
from typing import Callable, Optional

import torch

from ... import ops

# live decode engines (weak): release_graphs() destroys their captured
# hipGraphs DETERMINISTICALLY instead of leaving destruction to GC, whose
# graph-pool frees can interleave with a later context's allocations
# (observed mixed-suite segfault, NOTES_ROUND2 "Known issue")
_ENGINES = weakref.WeakSet()


def release_graphs():
    """Destroy all captured decode graphs + engine buffers now (ordered,
    synchronized).  Safe to call any time; engines re-capture on next use."""
    if not torch.cuda.is_available():
        return
    torch.cuda.synchronize()
    for engine in list(_ENGINES):
        g = getattr(engine, "graph", None)
        if g is not None:
            try:
                g.reset()
            except (RuntimeError, AttributeError):
                pass
            engine.graph = None
        model = getattr(engine, "model", None)
        if model is not None and getattr(model, "_decode_engine", None) is engine:
            model._decode_engine = None
    torch.cuda.synchronize()


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
    use_graph: bool = True
    # shaping fns made of pure torch ops (ILQL heads, logit masks) may be
    # captured into the decode graph
    graph_safe_shaping: bool = False

    @classmethod
    def from_kwargs(cls, **kwargs) -> "GenerateConfig":
        kwargs = dict(kwargs)
        kwargs.pop("max_length", None)
        known = {f for f in cls.__dataclass_fields__}
        unknown = [k for k in kwargs if k not in known]
        if unknown:
            # HF generate kwargs we do not implement (num_beams,
            # repetition_penalty, ...) must not be dropped silently: the
            # run would quietly sample a different distribution
            import warnings

            warnings.warn(
                f"gen_kwargs not supported by the native generate and IGNORED: {unknown} "
                f"(supported: {sorted(known)})", stacklevel=3)
        return cls(**{k: v for k, v in kwargs.items() if k in known})

    def sample_key(self):
        return (self.do_sample, self.temperature, self.top_k, self.top_p,
                self.eos_token_id, self.pad_token_id)


class FusedStageRefs:
    """Weight references + scratch buffers for the fused stage-kernel decode
    step (csrc/decode_mega.hip): 5 kernels per layer instead of ~12."""

    def __init__(self, model, batch: int, device):
        import torch.nn as nn

        from ...parallel import topo

        cfg = model.config
        base_ok = (
            not cfg.parallel_residual
            and cfg.num_heads == cfg.num_kv_heads
            and cfg.head_dim in (64, 128, 256)
            and cfg.position_encoding in ("learned", "rope")
            and not cfg.extra.get("pre_embed_norm")
            and not cfg.swiglu
            and cfg.hidden_size % 32 == 0
            and cfg.intermediate_size % 32 == 0
            and topo.tp_size() == 1
            and model.embed_norm is None
        )
        if not base_ok:
            raise ValueError("arch not fused-decode eligible")
        from ... import ops

        self.act_code = ops.ACT_CODES.get(cfg.activation)
        if self.act_code is None:
            raise ValueError(f"activation {cfg.activation} not fused-decode eligible")
        for layer in model.layers:
            for mod in (layer.attn.qkv_proj, layer.attn.o_proj, layer.mlp.fc_in,
                        layer.mlp.down_proj):
                if type(mod) is not nn.Linear or mod.weight.dtype != torch.bfloat16:
                    raise ValueError("fused decode needs plain bf16 linears")
        if model.lm_head.weight.dtype != torch.bfloat16:
            raise ValueError("fused decode needs a bf16 lm_head")

        H, I = cfg.hidden_size, cfg.intermediate_size
        L = cfg.num_layers
        self.cfg = cfg
        self.rms = cfg.norm == "rmsnorm"
        self.eps = cfg.norm_eps
        self.scale = model.layers[0].attn.scale
        self.rot = model.layers[0].attn.rot
        self.pos_offset = cfg.extra.get("position_offset", 0)
        self.wte = model.embed_tokens.weight
        self.wpe = model.embed_positions.weight if model.embed_positions is not None else None
        self.rcos, self.rsin = (model.rope_tables if cfg.position_encoding == "rope"
                                else (None, None))
        self.lnf_w = model.final_norm.weight
        self.lnf_b = model.final_norm.bias
        self.lm_w = model.lm_head.weight
        self.lm_b = model.lm_head.bias
        self.layers = []
        for layer in model.layers:
            self.layers.append(dict(
                ln1_w=layer.ln_1.weight, ln1_b=layer.ln_1.bias,
                qkv_w=layer.attn.qkv_proj.weight, qkv_b=layer.attn.qkv_proj.bias,
                o_w=layer.attn.o_proj.weight, o_b=layer.attn.o_proj.bias,
                ln2_w=layer.ln_2.weight, ln2_b=layer.ln_2.bias,
                fc_w=layer.mlp.fc_in.weight, fc_b=layer.mlp.fc_in.bias,
                down_w=layer.mlp.down_proj.weight, down_b=layer.mlp.down_proj.bias,
            ))
        # v3 (W-stationary, partial-stat slabs) needs 16-multiple batch and
        # 16-multiple widths; everything the flagship runs qualifies
        self.use_v3 = (batch in (16, 32, 64, 128, 256) and H % 128 == 0 and I % 128 == 0
                       and cfg.qkv_out % 16 == 0)
        bf = torch.bfloat16
        self.x = torch.empty(batch, H, dtype=bf, device=device)
        self.x1 = torch.empty(batch, H, dtype=bf, device=device)
        self.qkv = torch.empty(batch, cfg.qkv_out, dtype=bf, device=device)
        self.act = torch.empty(batch, I, dtype=bf, device=device)
        # partial-stat slabs [N/16 panels, B, 2]; embed fills slot 0 of ps_x
        # (nparts=1 for the first layer's qkv).  Sized to also cover the v1
        # fallback's 2L+1 per-norm slots.
        nslots = max(H // 16, 2 * L + 1, 1)
        self.ps_x = torch.zeros(nslots, batch, 2, dtype=torch.float32, device=device)
        self.ps_x1 = torch.zeros(nslots, batch, 2, dtype=torch.float32, device=device)
        self.stats = self.ps_x  # embed_stats writes slot 0 + zeroes the rest
        self.packed = torch.zeros(batch, dtype=torch.long, device=device)


class DecodeEngine:
    """Persistent hipGraph-captured decode step."""

    def __init__(self, model, batch: int, cache_len: int, max_new: int, gen: GenerateConfig,
                 device, shaping_fn=None):
        self.model = model
        self.shaping_fn = shaping_fn
        self.batch = batch
        self.cache_len = cache_len
        self.max_new = max_new
        self.gen = gen
        self.device = device
        self.seed = gen.seed if gen.seed is not None else int(torch.randint(0, 2**31 - 1, (1,)).item())
        self.kv = model.new_kv_cache(batch, cache_len, device=device)
        # fused stage-kernel step (5 kernels/layer): eligible archs only, and
        # only for the plain sampler the lm_sample kernel implements.
        # OPT-IN (TRLX_AMD_FUSED_DECODE=1): the same-box A/B currently favors
        # the module path (its in-graph hipBLASLt GEMMs run 2x faster than the
        # stage kernels — see profiles/ and tools/bench_stage_gemm.py); the
        # fused path stays in for iteration until it wins e2e.
        self.fused = None
        if (shaping_fn is None and os.environ.get("TRLX_AMD_FUSED_DECODE") == "1"
                and gen.top_k in (0, None) and gen.top_p in (1.0, None)
                and gen.min_new_tokens == 0):
            try:
                self.fused = FusedStageRefs(model, batch, device)
            except (ValueError, AttributeError):
                self.fused = None
        self.cur_tok = torch.zeros(batch, 1, dtype=torch.long, device=device)
        self.key_starts = torch.zeros(batch, dtype=torch.int32, device=device)
        self.seq_lens = torch.zeros(batch, dtype=torch.int32, device=device)
        self.pos_ids = torch.zeros(batch, 1, dtype=torch.int32, device=device)
        self.cache_idx = torch.zeros(1, dtype=torch.long, device=device)
        self.rng_offset = torch.zeros(1, dtype=torch.long, device=device)
        self.step_col = torch.zeros(1, dtype=torch.long, device=device)
        self.finished = torch.zeros(batch, dtype=torch.bool, device=device)
        pad = gen.pad_token_id if gen.pad_token_id is not None else (gen.eos_token_id or 0)
        self.pad_id = pad
        self.out_tokens = torch.full((batch, max_new), pad, dtype=torch.long, device=device)
        self.graph = None
        _ENGINES.add(self)

    def matches(self, batch, needed_cache, max_new, gen: GenerateConfig) -> bool:
        return (batch == self.batch and needed_cache <= self.cache_len
                and max_new <= self.max_new and gen.sample_key() == self.gen.sample_key())

    def _sample(self, logits):
        if not self.gen.do_sample:
            return logits.argmax(dim=-1)
        return ops.sample_token(logits, self.gen.temperature, self.gen.top_k, self.gen.top_p,
                                seed=self.seed, offset=self.rng_offset)

    def _advance(self, tok):
        """Post-sampling state updates (shared by graph step and prefill).

        One fused HIP kernel (csrc/decode_advance.hip) replaces ~11
        elementwise launches of bookkeeping per token; it also pre-computes
        the NEXT step's position ids and bumps seq_lens/cache_idx, so the
        step body is model + sample + advance and nothing else.  The torch
        fallback reproduces the same semantics op by op."""
        ext = ops._load_ext()
        if ext is not None and hasattr(ext, "decode_advance") and tok.is_cuda:
            ext.decode_advance(
                tok.contiguous(), self.out_tokens, self.cur_tok.view(-1), self.finished,
                self.rng_offset, self.step_col, self.cache_idx, self.seq_lens,
                self.pos_ids.view(-1), self.key_starts,
                -1 if self.gen.eos_token_id is None else self.gen.eos_token_id, self.pad_id,
            )
            return
        if self.gen.eos_token_id is not None:
            tok = torch.where(self.finished, torch.full_like(tok, self.pad_id), tok)
            self.finished |= tok == self.gen.eos_token_id
        self.out_tokens.index_copy_(1, self.step_col, tok.unsqueeze(1))
        self.cur_tok.copy_(tok.unsqueeze(1))
        self.rng_offset += 1
        self.step_col += 1
        self.cache_idx += 1
        self.seq_lens += 1
        self.pos_ids.copy_(
            (self.cache_idx - self.key_starts.to(torch.long)).to(torch.int32).unsqueeze(1))

    def _fused_step(self):
        """One decode token through the fused stage kernels
        (csrc/decode_mega.hip): embed+stats, then per layer
        [ln1-folded qkv GEMM] -> [fused decode attention] ->
        [o GEMM + residual + ln2 stats] -> [ln2-folded fc GEMM + act] ->
        [down GEMM + residual + next-ln1 stats], then the final-norm-folded
        lm_head GEMM fused with Gumbel-max sampling, then advance — 63
        launches/token for GPT-2 vs ~150 on the module path, each folding
        the norm/bias/act/residual elementwise work into a GEMM epilogue."""
        from ... import ops

        ext = ops._load_ext()
        f = self.fused
        cur = self.cur_tok.view(-1)
        pos = self.pos_ids.view(-1)
        ext.embed_stats(f.wte, f.wpe, cur, pos, f.pos_offset, f.x, f.stats, f.packed)
        temperature = self.gen.temperature if self.gen.do_sample else 0.0
        if f.use_v3:
            # v3 W-stationary consumers (qkv/fc/lm read each weight byte
            # ONCE) + v2 producers emitting partial-stat slabs for the next
            # norm.  nparts: the embed kernel wrote one full-row slot; each
            # down/o stage writes H/16 16-col partials.
            H16 = f.cfg.hidden_size // 16
            nparts_x = 1  # embed slot 0
            for li, lay in enumerate(f.layers):
                ext.stage_gemm_v3(f.x, lay["qkv_w"], lay["qkv_b"], f.qkv, f.ps_x, nparts_x,
                                  lay["ln1_w"], lay["ln1_b"], f.rms, f.eps, 0, None)
                attn = ext.fused_decode_attention(
                    f.qkv, self.kv.k[li], self.kv.v[li], self.seq_lens, self.key_starts,
                    f.rcos, f.rsin, self.cache_idx, f.rot, f.cfg.rope_interleaved, f.scale,
                ).view(self.batch, -1)
                ext.stage_gemm_v2(attn, lay["o_w"], lay["o_b"], f.x1, False,
                                  None, None, False, 0.0, 0, f.x, f.ps_x1)
                ext.stage_gemm_v3(f.x1, lay["fc_w"], lay["fc_b"], f.act, f.ps_x1, H16,
                                  lay["ln2_w"], lay["ln2_b"], f.rms, f.eps, f.act_code, None)
                ext.stage_gemm_v2(f.act, lay["down_w"], lay["down_b"], f.x, False,
                                  None, None, False, 0.0, 0, f.x1, f.ps_x)
                nparts_x = H16
            ext.lm_sample_v3(f.x, f.lm_w, f.lm_b, f.ps_x, nparts_x, f.lnf_w, f.lnf_b,
                             f.packed, f.rms, f.eps, temperature, self.seed, self.rng_offset)
            ext.advance_packed(f.packed, self.out_tokens, cur, self.finished,
                               self.rng_offset, self.step_col, self.cache_idx,
                               self.seq_lens, pos, self.key_starts,
                               -1 if self.gen.eos_token_id is None else self.gen.eos_token_id,
                               self.pad_id)
            return
        for li, lay in enumerate(f.layers):
            ext.stage_gemm(f.x, lay["qkv_w"], lay["qkv_b"], f.qkv, f.stats[2 * li],
                           lay["ln1_w"], lay["ln1_b"], f.rms, f.eps, 0, None, None)
            attn = ext.fused_decode_attention(
                f.qkv, self.kv.k[li], self.kv.v[li], self.seq_lens, self.key_starts,
                f.rcos, f.rsin, self.cache_idx, f.rot, f.cfg.rope_interleaved, f.scale,
            ).view(self.batch, -1)
            ext.stage_gemm(attn, lay["o_w"], lay["o_b"], f.x1, None, None, None,
                           False, 0.0, 0, f.x, f.stats[2 * li + 1])
            ext.stage_gemm(f.x1, lay["fc_w"], lay["fc_b"], f.act, f.stats[2 * li + 1],
                           lay["ln2_w"], lay["ln2_b"], f.rms, f.eps, f.act_code, None, None)
            ext.stage_gemm(f.act, lay["down_w"], lay["down_b"], f.x, None, None, None,
                           False, 0.0, 0, f.x1, f.stats[2 * li + 2])
        ext.lm_sample(f.x, f.lm_w, f.lm_b, f.stats[2 * len(f.layers)], f.lnf_w, f.lnf_b,
                      f.packed, f.rms, f.eps, temperature, self.seed, self.rng_offset)
        ext.advance_packed(f.packed, self.out_tokens, cur, self.finished, self.rng_offset,
                           self.step_col, self.cache_idx, self.seq_lens, pos, self.key_starts,
                           -1 if self.gen.eos_token_id is None else self.gen.eos_token_id,
                           self.pad_id)

    def _step(self):
        """One decode token — everything device-side (hipGraph body).

        State convention: on entry, seq_lens/cache_idx/pos_ids already
        describe THIS step (the previous _advance set them)."""
        if self.fused is not None:
            return self._fused_step()
        out = self.model(
            self.cur_tok, kv_cache=self.kv, position_ids=self.pos_ids, seq_lens=self.seq_lens,
            key_starts=self.key_starts, cache_idx=self.cache_idx, start_pos=0,
            return_logits=False,
        )
        logits = self.model.lm_head(out.last_hidden_state[:, -1:, :])[:, 0]
        if self.shaping_fn is not None:
            logits = self.shaping_fn(logits.float(), out.last_hidden_state[:, -1],
                                     self.cur_tok[:, 0])
        # bf16 logits go straight to the sampler (same values the fp32 cast
        # would carry — the lm_head output is already bf16-rounded)
        tok = self._sample(logits)
        self._advance(tok)

    def _scratch_state(self):
        """In-bounds dummy state for warmup/capture runs (cache row 0 gets
        scribbled; the real prefill overwrites it afterwards)."""
        self.step_col.zero_()
        self.cache_idx.zero_()
        self.seq_lens.fill_(1)
        self.pos_ids.zero_()
        self.rng_offset.zero_()
        self.finished.zero_()

    def capture(self):
        torch.cuda.synchronize()
        side = torch.cuda.Stream()
        side.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(side):
            for _ in range(2):
                self._scratch_state()
                self._step()
        torch.cuda.current_stream().wait_stream(side)
        torch.cuda.synchronize()
        self._scratch_state()
        self.graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self.graph):
            self._step()

    def run(self, input_ids, attention_mask, max_new_tokens: int) -> torch.Tensor:
        B, T = input_ids.shape
        # capture happens BEFORE prefill: warmup scribbles on cache row 0
        if self.graph is None and max_new_tokens > 1:
            self.capture()
        if os.environ.get("TRLX_AMD_FP8_DECODE") == "1":
            # training may have updated the weights since the last generate
            # (FusedAdamW writes arenas kernel-side, invisible to version
            # counters) — requantize into the SAME buffers the graph reads
            ops.refresh_fp8_caches(self.model)

        # reset per-call state.  The fused advance will bump cache_idx/
        # seq_lens and derive pos_ids for the first replay, so they start
        # one step "behind": after the prefill-token advance the first
        # replay sees seq_lens=T+1, cache_idx=T, pos=T-key_starts (the same
        # values the pre-fusion engine computed inside its step).
        self.key_starts.copy_((T - attention_mask.sum(-1)).to(torch.int32))
        self.seq_lens.fill_(T)
        self.cache_idx.fill_(T - 1)
        self.step_col.zero_()
        if self.gen.seed is not None:
            # explicit seed: same (seed, inputs) -> same tokens, like eager.
            # With no seed the offset keeps advancing across calls so reused
            # engines stay fresh (the capture bakes the random seed in).
            self.rng_offset.zero_()
        self.finished.zero_()
        self.out_tokens.fill_(self.pad_id)

        # eager prefill fills cache[0:T] and yields the first sampled token
        out = self.model(input_ids, attention_mask=attention_mask, kv_cache=self.kv,
                         start_pos=0, return_logits=False)
        logits = self.model.lm_head(out.last_hidden_state[:, -1:, :])[:, 0].float()
        if self.shaping_fn is not None:
            logits = self.shaping_fn(logits, out.last_hidden_state[:, -1], input_ids[:, -1])
        tok = self._sample(logits)
        self._advance(tok)

        eos = self.gen.eos_token_id
        done_steps = 1
        for i in range(max_new_tokens - 1):
            self.graph.replay()
            done_steps += 1
            if eos is not None and (i + 1) % 16 == 0 and bool(self.finished.all()):
                break
        return torch.cat([input_ids, self.out_tokens[:, :done_steps]], dim=1)


def _graphs_enabled() -> bool:
    return os.environ.get("TRLX_AMD_NO_GRAPHS") != "1"


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

    try:
        shaping_graphable = shaping_fn is None or gen.graph_safe_shaping
        has_virtual = getattr(model, "num_virtual_tokens", 0) > 0
        alibi = getattr(model.config, "position_encoding", None) == "alibi"
        # the graph engine needs the fused decode kernels (device-side cache
        # index); unsupported head dims use the eager loop's host-side state
        head_ok = getattr(model.config, "head_dim", 64) in (32, 64, 128, 256)
        if (device.type == "cuda" and shaping_graphable and gen.use_graph and not alibi
                and head_ok and not has_virtual
                and _graphs_enabled() and gen.max_new_tokens > 1 and gen.min_new_tokens == 0):
            engine = getattr(model, "_decode_engine", None)
            needed = T + gen.max_new_tokens
            if (engine is None or not engine.matches(B, needed, gen.max_new_tokens, gen)
                    or engine.shaping_fn is not shaping_fn):
                engine = DecodeEngine(model, B, needed, gen.max_new_tokens, gen, device,
                                      shaping_fn=shaping_fn)
                model._decode_engine = engine
            return engine.run(input_ids, attention_mask, gen.max_new_tokens)
        return _generate_eager(model, input_ids, attention_mask, gen, shaping_fn)
    finally:
        if was_training:
            model.train()


def _generate_eager(model, input_ids, attention_mask, gen: GenerateConfig, shaping_fn):
    """Reference loop: CPU, shaping fns, graphs disabled, or PEFT virtual
    tokens (the prefill inserts nv prompt/prefix slots into the cache; all
    decode positions shift by nv)."""
    B, T = input_ids.shape
    device = input_ids.device
    nv = getattr(model, "num_virtual_tokens", 0)
    key_starts = (T - attention_mask.sum(-1)).to(torch.int32)
    kv = model.new_kv_cache(B, T + nv + gen.max_new_tokens, device=device)

    out = model(input_ids, attention_mask=attention_mask, kv_cache=kv, start_pos=0,
                return_logits=False)
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
        start_pos = T + nv + step
        position_ids = (start_pos - key_starts).to(torch.int32).unsqueeze(1)
        seq_lens = torch.full((B,), start_pos + 1, dtype=torch.int32, device=device)
        # on GPU route through the same fused decode kernels as the graph
        # engine (identical numerics)
        cache_idx = None
        if device.type == "cuda" and ops.extension_available():
            cache_idx = torch.full((1,), start_pos, dtype=torch.long, device=device)
        out = model(
            next_tok.unsqueeze(1), kv_cache=kv, start_pos=start_pos, position_ids=position_ids,
            seq_lens=seq_lens, key_starts=key_starts, return_logits=False, cache_idx=cache_idx,
        )
        logits_last = model.lm_head(out.last_hidden_state[:, -1:, :])[:, 0]
        hidden_last = out.last_hidden_state[:, -1]

    if not generated:
        return input_ids
    return torch.cat([input_ids, torch.stack(generated, dim=1)], dim=1)
