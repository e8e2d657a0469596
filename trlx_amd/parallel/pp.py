"""Pipeline parallelism: stage-split transformer + fill-drain schedule.

Parity target: the reference's NeMo/Apex pipeline path (SURVEY.md §2.2 PP
row: per-stage model construction modeling_nemo_ppo.py:497-536, p2p
activations of shape [seq, micro_batch, hidden], fwd/bwd schedule 713-731).

MI355X design: activations/grads move between stages as [B, T, H] bf16
tensors over RCCL p2p (xGMI on-node).  Default schedule is 1F1B (warmup
forwards, one-forward/one-backward steady state, drain — activation
footprint bounded by pipeline depth); GPipe fill-drain is kept as an
alternative.  Token-level pipelined generation is a planned follow-up.

Rank layout (topo.init_model_parallel): pp is the middle axis —
rank = (dp_idx * pp_size + pp_idx) * tp_size + tp_idx — so a PP group's
stages sit tp_size apart (still intra-node for tp*pp <= 8).
"""

from typing import List, Optional

import torch
import torch.distributed as dist
import torch.nn as nn

from . import topo
from ..models.nn.config import TransformerConfig
from ..models.nn.transformer import Block, Norm, TransformerOutput


def split_layers(num_layers: int, pp: int, stage: int):
    """Contiguous layer range for ``stage`` (early stages get the remainder)."""
    base = num_layers // pp
    extra = num_layers % pp
    lo = stage * base + min(stage, extra)
    hi = lo + base + (1 if stage < extra else 0)
    return lo, hi


class PipelineStage(nn.Module):
    """One pipeline stage of the causal transformer.

    Stage 0 owns the embeddings; the last stage owns the final norm +
    lm_head.  Weights load from a full state dict via ``load_full_state_dict``
    (the PP resharding of reference modeling_nemo_ppo.py:321-352).
    """

    def __init__(self, config: TransformerConfig, stage: int, num_stages: int):
        super().__init__()
        self.config = config
        self.stage = stage
        self.num_stages = num_stages
        self.is_first = stage == 0
        self.is_last = stage == num_stages - 1
        self.lo, self.hi = split_layers(config.num_layers, num_stages, stage)
        cfg = config

        if self.is_first:
            self.embed_tokens = nn.Embedding(cfg.vocab_size, cfg.hidden_size)
            if cfg.position_encoding == "learned":
                off = cfg.extra.get("position_offset", 0)
                self.embed_positions = nn.Embedding(cfg.max_position_embeddings + off, cfg.hidden_size)
            else:
                self.embed_positions = None
        # LOCAL layer indices: Block.layer_idx is only the KV-cache slot, and
        # each stage owns its own cache of len(self.layers) entries
        self.layers = nn.ModuleList(Block(cfg, i - self.lo) for i in range(self.lo, self.hi))
        # set to a local layer index to stash the true residual stream fed
        # into that layer (consumed by the hydra frozen branch on the last
        # stage); read back from ``self.last_stash`` right after forward()
        self.stash_local_layer: Optional[int] = None
        self.last_stash = None
        if self.is_last:
            self.final_norm = Norm(cfg)
            self.lm_head = nn.Linear(cfg.hidden_size, cfg.vocab_size, bias=cfg.lm_head_bias)
            if cfg.tie_word_embeddings and self.is_first:
                self.lm_head.weight = self.embed_tokens.weight
        from ..ops.reference import rope_cos_sin as _rcs

        if cfg.position_encoding == "rope":
            rot = int(cfg.head_dim * cfg.rope_pct)
            rot -= rot % 2
            cos, sin = _rcs(cfg.max_position_embeddings, rot, cfg.rope_base)
            self.register_buffer("rope_cos", cos, persistent=False)
            self.register_buffer("rope_sin", sin, persistent=False)
        else:
            self.rope_cos = self.rope_sin = None

    @property
    def rope_tables(self):
        return (self.rope_cos, self.rope_sin) if self.rope_cos is not None else None

    def load_full_state_dict(self, sd: dict):
        """Filter + reindex a full-model state dict onto this stage."""
        own = {}
        for k, v in sd.items():
            if k.startswith("layers."):
                idx = int(k.split(".")[1])
                if self.lo <= idx < self.hi:
                    own[f"layers.{idx - self.lo}." + k.split(".", 2)[2]] = v
            elif k.startswith(("embed_tokens.", "embed_positions.")) and self.is_first:
                own[k] = v
            elif k.startswith(("final_norm.", "lm_head.")) and self.is_last:
                own[k] = v
        missing, unexpected = self.load_state_dict(own, strict=False)
        return missing, unexpected

    def make_context(self, input_ids, attention_mask):
        from ..models.nn.transformer import AttentionContext

        B, T = input_ids.shape[:2]
        device = input_ids.device
        key_starts = None
        if attention_mask is not None:
            mask = attention_mask.to(torch.int32)
            position_ids = (mask.cumsum(-1) - 1).clamp(min=0).to(torch.int32)
            key_starts = (T - mask.sum(-1)).to(torch.int32)
        else:
            position_ids = (torch.arange(T, device=device, dtype=torch.int32)
                            .unsqueeze(0).expand(B, T).contiguous())
        return AttentionContext(position_ids=position_ids, key_starts=key_starts, start_pos=0)

    def forward(self, hidden_or_ids, ctx, kv_cache=None):
        if self.is_first:
            h = self.embed_tokens(hidden_or_ids)
            if self.embed_positions is not None:
                off = self.config.extra.get("position_offset", 0)
                h = h + self.embed_positions(ctx.position_ids.long() + off)
        else:
            h = hidden_or_ids
        res = None
        self.last_stash = None
        for i, layer in enumerate(self.layers):
            if self.stash_local_layer is not None and i == self.stash_local_layer:
                self.last_stash = h if res is None else h + res
            h, res = layer(h, ctx, self.rope_tables, kv_cache, res=res)
        if self.stash_local_layer is not None and self.stash_local_layer == len(self.layers):
            self.last_stash = h if res is None else h + res
        if res is not None:
            # stage boundaries exchange the materialized stream
            h = h + res
        if self.is_last:
            h = self.final_norm(h)
        return h

    def project(self, h):
        assert self.is_last
        return self.lm_head(h)

    def new_kv_cache(self, batch: int, max_len: int, device=None, dtype=None):
        from ..models.nn.transformer import KVCache

        p = next(self.parameters())
        kv_heads_local = self.config.num_kv_heads // topo.tp_size()
        return KVCache(len(self.layers), batch, kv_heads_local, max_len,
                       self.config.head_dim, device or p.device, dtype or p.dtype)


class PipelineRunner:
    """Pipeline forward/backward over the PP group.

    Two schedules:
    - ``1f1b`` (default; parity: the NeMo/Megatron schedule): each stage runs
      ``pp_size - idx - 1`` warmup forwards, then alternates one-forward/
      one-backward (backwards in FIFO order), then drains.  Peak live
      activations are bounded by the pipeline depth instead of the microbatch
      count.  Sends are non-blocking (isend) so the crossing act/grad
      messages of the steady state can't deadlock.
    - ``gpipe``: plain fill-drain (all forwards, then all backwards LIFO).
    """

    def __init__(self, stage: PipelineStage, pp_group=None, pp_ranks: Optional[List[int]] = None,
                 schedule: str = "1f1b"):
        self.stage = stage
        self.group = pp_group
        self.schedule = schedule
        # global ranks of the pipeline stages in order
        if pp_ranks is None:
            pp_ranks = list(range(dist.get_world_size()))
        self.pp_ranks = pp_ranks
        self.idx = pp_ranks.index(dist.get_rank())
        self.prev = pp_ranks[self.idx - 1] if self.idx > 0 else None
        self.next = pp_ranks[self.idx + 1] if self.idx < len(pp_ranks) - 1 else None
        self._pending = []  # (work, tensor) — tensor kept alive until waited

    def _send(self, t: torch.Tensor, dst: int):
        t = t.contiguous()
        self._pending.append((dist.isend(t, dst), t))

    def _drain_sends(self):
        for work, _ in self._pending:
            work.wait()
        self._pending.clear()

    def _recv(self, shape, dtype, device, src: int) -> torch.Tensor:
        t = torch.empty(shape, dtype=dtype, device=device)
        dist.recv(t, src)
        return t

    def forward_backward(self, microbatches, loss_fn, hidden_dtype=torch.float32):
        """Run pipelined fwd+bwd.  ``microbatches``: list of dicts with
        input_ids/attention_mask (used on stage 0 for embeddings and on every
        stage for the attention context).  ``loss_fn(h_last, mb)`` is
        evaluated on the last stage with the POST-final-norm hidden state
        (callers project to logits/values themselves); it returns a loss or a
        ``(loss, stats_dict)`` pair.  Returns mean loss (last stage) or None;
        per-microbatch stats land in ``self.last_stats``.  Gradients
        accumulate into stage parameters (summed over microbatches, like the
        single-process accumulation path)."""
        self.last_stats = []
        if self.schedule == "1f1b":
            return self._forward_backward_1f1b(microbatches, loss_fn, hidden_dtype)
        return self._forward_backward_gpipe(microbatches, loss_fn, hidden_dtype)

    def _eval_loss(self, h, mb, loss_fn, losses):
        out = loss_fn(h, mb)
        if isinstance(out, tuple):
            loss, stats = out
            self.last_stats.append(stats)
        else:
            loss = out
        losses.append(loss.detach())
        return loss

    @torch.no_grad()
    def forward_inference(self, input_ids, attention_mask=None, kv_cache=None,
                          ctx=None, hidden_dtype=torch.float32):
        """One no-grad chained forward through the pipeline (experience phase,
        generation prefill/decode).  Every PP rank calls this with the SAME
        input_ids; returns the post-final-norm hidden on the last stage, None
        elsewhere."""
        stage = self.stage
        device = next(stage.parameters()).device
        ids = input_ids.to(device)
        if ctx is None:
            mask = attention_mask.to(device) if attention_mask is not None else None
            ctx = stage.make_context(ids, mask)
        if stage.is_first:
            inp = ids
        else:
            B, T = ids.shape[:2]
            inp = self._recv((B, T, stage.config.hidden_size), hidden_dtype, device, self.prev)
        out = stage(inp, ctx, kv_cache=kv_cache)
        if not stage.is_last:
            self._send(out, self.next)
            self._drain_sends()
            return None
        return out

    def _forward_backward_1f1b(self, microbatches, loss_fn, hidden_dtype=torch.float32):
        stage = self.stage
        H = stage.config.hidden_size
        device = next(stage.parameters()).device
        from collections import deque

        stashes = deque()
        losses = []
        n = len(microbatches)
        fwd_i = 0

        def do_forward():
            nonlocal fwd_i
            mb = microbatches[fwd_i]
            fwd_i += 1
            ids = mb["input_ids"].to(device)
            mask = mb.get("attention_mask")
            mask = mask.to(device) if mask is not None else None
            ctx = stage.make_context(ids, mask)
            if stage.is_first:
                h_in = None
                inp = ids
            else:
                B, T = ids.shape
                h_in = self._recv((B, T, H), hidden_dtype, device, self.prev)
                h_in.requires_grad_(True)
                inp = h_in
            out = stage(inp, ctx)
            if stage.is_last:
                loss = self._eval_loss(out, mb, loss_fn, losses)
                stashes.append((h_in, loss))
            else:
                self._send(out.detach(), self.next)
                stashes.append((h_in, out))

        def do_backward():
            h_in, out_or_loss = stashes.popleft()  # FIFO: 1F1B backs up in order
            if stage.is_last:
                out_or_loss.backward()
            else:
                grad = self._recv(out_or_loss.shape, hidden_dtype, device, self.next)
                if out_or_loss.requires_grad:
                    out_or_loss.backward(grad)
                # else: fully frozen first stage — grad is dropped
            if not stage.is_first:
                self._send(h_in.grad, self.prev)

        warmup = min(len(self.pp_ranks) - self.idx - 1, n)
        for _ in range(warmup):
            do_forward()
        for _ in range(n - warmup):
            do_forward()
            do_backward()
        for _ in range(warmup):
            do_backward()
        self._drain_sends()
        self._sync_tied_embedding_grads()
        if losses:
            return torch.stack(losses).mean()
        return None

    def _sync_tied_embedding_grads(self):
        """Tied input/output embeddings live on DIFFERENT stages under PP:
        stage 0 holds embed_tokens, the last stage holds lm_head (initialized
        from the same weight).  Their grads must be SUMMED so both copies
        take the same update and stay tied (the Megatron/NeMo embedding-group
        all-reduce; reference megatron stack).  Exchange over p2p between the
        two end stages."""
        stage = self.stage
        if not stage.config.tie_word_embeddings or len(self.pp_ranks) < 2:
            return
        if stage.is_first and stage.is_last:
            return
        # frozen tied pairs (e.g. num_layers_unfrozen freezing the embeddings
        # AND the tied lm_head) take no updates — nothing to synchronize
        if stage.is_first and not stage.embed_tokens.weight.requires_grad:
            return
        if stage.is_last and not stage.lm_head.weight.requires_grad:
            return
        first, last = self.pp_ranks[0], self.pp_ranks[-1]
        if stage.is_first:
            g = stage.embed_tokens.weight.grad
            if g is None:
                g = torch.zeros_like(stage.embed_tokens.weight)
            dist.send(g.contiguous(), last)
            total = torch.empty_like(g)
            dist.recv(total, last)
            stage.embed_tokens.weight.grad = total
        elif stage.is_last:
            other = torch.empty_like(stage.lm_head.weight)
            dist.recv(other, first)
            g = stage.lm_head.weight.grad
            total = other if g is None else other + g
            dist.send(total.contiguous(), first)
            stage.lm_head.weight.grad = total

    def _forward_backward_gpipe(self, microbatches, loss_fn, hidden_dtype=torch.float32):
        stage = self.stage
        H = stage.config.hidden_size
        device = next(stage.parameters()).device
        stashes = []
        losses = []

        # ---- fill: forwards ------------------------------------------------
        for mb in microbatches:
            ids = mb["input_ids"].to(device)
            mask = mb.get("attention_mask")
            mask = mask.to(device) if mask is not None else None
            ctx = stage.make_context(ids, mask)
            if stage.is_first:
                inp = ids
                h_in = None
            else:
                B, T = ids.shape
                h_in = self._recv((B, T, H), hidden_dtype, device, self.prev)
                h_in.requires_grad_(True)
                inp = h_in
            out = stage(inp, ctx)
            if stage.is_last:
                loss = self._eval_loss(out, mb, loss_fn, losses)
                stashes.append((h_in, loss))
            else:
                self._send(out.detach(), self.next)
                stashes.append((h_in, out))

        # ---- drain: backwards (reverse order) ------------------------------
        for h_in, out_or_loss in reversed(stashes):
            if stage.is_last:
                out_or_loss.backward()
            else:
                grad = self._recv(out_or_loss.shape, hidden_dtype, device, self.next)
                if out_or_loss.requires_grad:
                    out_or_loss.backward(grad)
                # else: fully frozen first stage — grad is dropped
            if not stage.is_first:
                self._send(h_in.grad, self.prev)

        self._drain_sends()
        self._sync_tied_embedding_grads()
        if losses:
            return torch.stack([l.detach() for l in losses]).mean()
        return None
