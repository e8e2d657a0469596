"""Pipeline-parallel RL model: the PP-sharded native transformer with the
value head and the hydra frozen reference branch living on the LAST stage.

Parity target: the reference's NeMo/Megatron pipeline path — per-stage model
construction (reference modeling_nemo_ppo.py:497-536: value head + reference
heads only on the ``post_process`` stage), the pipelined train step driven by
the trainer (713-731), token-level pipelined generation (1158-1222), and the
TP×PP sharded checkpoint format (445-467 ``mp_rank_XX``; Megatron uses
``mp_rank_XX_YYY`` for TP×PP — adopted here).

MI355X design notes:
- activations cross stages as [B, T, H] tensors over RCCL p2p (xGMI on-node);
- the hydra reference branch re-runs only the top ``num_layers_unfrozen``
  blocks of the LAST stage from a stashed hidden state — one pipeline pass
  yields policy logprobs, values AND reference logprobs (the reference's NeMo
  path needs two pipeline passes + a CPU weight swap,
  modeling_nemo_ppo.py:1095-1156);
- decode generation is token-level: each new token makes one trip down the
  pipeline (per-stage KV caches) and the sampled id is broadcast back over
  the PP group, exactly the reference's NeMo decode structure.
"""

import json
import os
from typing import Dict, Optional

import torch
import torch.distributed as dist
import torch.nn as nn

from .. import ops
from ..parallel import topo
from ..parallel.pp import PipelineRunner, PipelineStage
from ..utils import logging
from ..utils.modeling import logprobs_of_labels, make_head
from .modeling_ppo import FrozenBranch
from .nn.config import PRESETS, TransformerConfig, preset
from .nn.generation import GenerateConfig
from .nn.transformer import AttentionContext

logger = logging.get_logger(__name__)


class PipelinedPPOModel(nn.Module):
    """PP-sharded causal LM + value head + hydra frozen branch (last stage).

    Built AFTER ``topo.init_model_parallel`` so the stage split follows the
    live topology; TP shards materialize inside each Block automatically.
    """

    is_pipelined = True

    def __init__(self, config: TransformerConfig, num_layers_unfrozen: int = -1,
                 num_value_layers_unfrozen: int = 0, peft_config=None,
                 with_value_head: bool = True):
        super().__init__()
        assert topo.pp_size() > 1, "PipelinedPPOModel requires pipeline_parallel_size > 1"
        assert peft_config is None, "peft + pipeline parallelism is not supported yet"
        assert num_value_layers_unfrozen == 0, \
            "num_value_layers_unfrozen > 0 + pipeline parallelism is not supported yet"
        assert config.position_encoding != "alibi", \
            "ALiBi models are not supported under pipeline parallelism yet"
        self.config = config
        self.peft_config = None
        self.num_layers_unfrozen = num_layers_unfrozen
        self.stage = PipelineStage(config, topo.pp_rank(), topo.pp_size())
        self.runner = PipelineRunner(self.stage, topo.pp_group(), topo.state().pp_ranks)
        self._sync_tied_embeddings()
        self.v_head = None
        self.frozen_head = None
        if self.stage.is_last:
            if with_value_head:
                self.v_head = make_head(config.hidden_size, 1, dtype=torch.float32)
            if num_layers_unfrozen > 0:
                n_local = len(self.stage.layers)
                if num_layers_unfrozen > n_local:
                    raise ValueError(
                        f"num_layers_unfrozen={num_layers_unfrozen} exceeds the last "
                        f"stage's {n_local} layers (pp={topo.pp_size()}); the hydra "
                        f"reference branch must fit on the last stage — lower "
                        f"num_layers_unfrozen or the pipeline depth")
                # FrozenBranch reads .layers/.final_norm/.lm_head/.config —
                # the stage provides all of them on the last stage
                self.frozen_head = FrozenBranch(self.stage, num_layers_unfrozen)
                self.stage.stash_local_layer = n_local - num_layers_unfrozen

    def _sync_tied_embeddings(self):
        """Tied word embeddings live on DIFFERENT stages under PP (stage 0:
        embed_tokens, last stage: an independent lm_head parameter) — copy the
        first stage's weight into the last stage's lm_head so the pair starts
        tied; the runner's tied-grad sync keeps them tied afterwards.  Must
        run BEFORE the frozen branch snapshots lm_head."""
        if not self.config.tie_word_embeddings or not dist.is_initialized():
            return
        st = self.stage
        if st.is_first and st.is_last:
            return
        ranks = self.runner.pp_ranks
        if st.is_first:
            dist.send(st.embed_tokens.weight.data.contiguous(), ranks[-1])
        elif st.is_last:
            buf = torch.empty_like(st.lm_head.weight.data)
            dist.recv(buf, ranks[0])
            st.lm_head.weight.data.copy_(buf)

    # --- construction helpers ------------------------------------------------

    @classmethod
    def from_any(cls, path_or_config, num_layers_unfrozen: int = -1, **kwargs):
        """Build from a TransformerConfig, a local HF dir (full weights loaded
        and resharded onto this stage), or a known preset name."""
        heads_sd = {}
        full_sd = None
        if isinstance(path_or_config, TransformerConfig):
            cfg = path_or_config
        elif isinstance(path_or_config, str) and os.path.isdir(path_or_config):
            from .nn.convert import load_hf_dir

            cfg, full_sd = load_hf_dir(path_or_config)
            heads_path = os.path.join(path_or_config, "wrapper_heads.pt")
            if os.path.exists(heads_path):
                heads_sd = torch.load(heads_path, map_location="cpu", weights_only=True)
        elif isinstance(path_or_config, str) and path_or_config in PRESETS:
            logger.warning(
                f"'{path_or_config}' is not a local directory; building a randomly-"
                f"initialized preset (no network access).")
            cfg = preset(path_or_config)
        else:
            raise OSError(f"cannot build a pipelined model from {path_or_config!r}")
        model = cls(cfg, num_layers_unfrozen=num_layers_unfrozen, **kwargs)
        if full_sd is not None:
            model.load_full_base_state_dict(full_sd)
        if heads_sd and model.stage.is_last:
            model.load_state_dict({k: v for k, v in heads_sd.items()
                                   if k.split(".")[0] in ("v_head", "frozen_head")},
                                  strict=False)
        return model

    def load_full_base_state_dict(self, sd: Dict[str, torch.Tensor]):
        """Reshard a FULL base-model state dict onto this stage; the frozen
        branch re-copies its blocks from the freshly loaded stage weights."""
        sd = {k[len("base_model."):] if k.startswith("base_model.") else k: v
              for k, v in sd.items()}
        if topo.tp_size() > 1:
            from ..parallel.tp import shard_state_dict_tp

            sd = shard_state_dict_tp(sd, self.config, topo.tp_rank(), topo.tp_size())
        self.stage.load_full_state_dict(sd)
        if self.frozen_head is not None:
            k = self.num_layers_unfrozen
            for dst, src in zip(self.frozen_head.blocks, self.stage.layers[-k:]):
                dst.load_state_dict(src.state_dict())
            self.frozen_head.final_norm.load_state_dict(self.stage.final_norm.state_dict())
            self.frozen_head.lm_head.load_state_dict(self.stage.lm_head.state_dict())

    def freeze_bottom(self, num_layers_unfrozen: int):
        """Freeze everything below the global top ``num_layers_unfrozen``
        blocks (embeddings included); matches
        utils.modeling.freeze_bottom_causal_layers.  With tied embeddings the
        last stage's lm_head copy is frozen too (the single-process model's
        shared tensor would be)."""
        if num_layers_unfrozen == -1:
            return
        cut = self.config.num_layers - num_layers_unfrozen  # first trainable layer
        st = self.stage
        if st.is_first:
            st.embed_tokens.weight.requires_grad_(False)
            if getattr(st, "embed_positions", None) is not None:
                st.embed_positions.weight.requires_grad_(False)
        for i, layer in enumerate(st.layers):
            if st.lo + i < cut:
                for p in layer.parameters():
                    p.requires_grad_(False)
        if st.is_last and self.config.tie_word_embeddings:
            st.lm_head.weight.requires_grad_(False)

    def cast_compute(self, dtype):
        self.to(dtype)
        if self.stage.rope_cos is not None:
            self.stage.rope_cos = self.stage.rope_cos.float()
            self.stage.rope_sin = self.stage.rope_sin.float()
        return self

    @property
    def hidden_dtype(self):
        return next(self.stage.parameters()).dtype

    @property
    def device(self):
        return next(self.stage.parameters()).device

    @property
    def base_model(self):
        # the trainer probes .base_model for freezing; the PP model freezes
        # through freeze_bottom instead
        return self.stage

    def num_parameters(self) -> int:
        return sum(p.numel() for p in self.parameters())

    # --- experience ----------------------------------------------------------

    @torch.no_grad()
    def forward_experience(self, input_ids, attention_mask, lo: int, hi: int,
                           labels: torch.Tensor):
        """Pipelined experience pass: ONE trip through the stages yields
        per-token policy logprobs, reference logprobs (hydra branch) and
        values for positions [lo, hi); results broadcast over the PP group so
        every stage returns identical tensors (the trainer's store must hold
        the same rollouts on all PP peers)."""
        st = self.stage
        device = self.device
        input_ids = input_ids.to(device)
        attention_mask = attention_mask.to(device) if attention_mask is not None else None
        h = self.runner.forward_inference(input_ids, attention_mask,
                                          hidden_dtype=self.hidden_dtype)
        B = input_ids.shape[0]
        W = hi - lo
        pack = torch.empty(3, B, W, dtype=torch.float32, device=device)
        if st.is_last:
            hs = h[:, lo:hi].contiguous()
            labels = labels.to(device)
            values = self.v_head(hs.to(self.v_head[0].weight.dtype)).squeeze(-1).float()
            logprobs = self._logprobs(st.lm_head, hs, labels)
            if self.frozen_head is not None:
                ctx = st.make_context(input_ids, attention_mask)
                rh = self.frozen_head.forward_hidden(st.last_stash, ctx, st.rope_tables,
                                                     logits_slice=(lo, hi))
                ref_logprobs = self._logprobs(self.frozen_head.lm_head, rh.contiguous(), labels)
            else:
                ref_logprobs = logprobs
            pack[0], pack[1], pack[2] = logprobs, ref_logprobs, values
        if dist.is_initialized():
            dist.broadcast(pack, src=self.runner.pp_ranks[-1], group=topo.pp_group())
        return pack[0], pack[1], pack[2]

    def _logprobs(self, lm_head, h, labels):
        """Fused hand-MFMA logprob path when eligible, library path otherwise
        (same eligibility as modeling_ppo.forward_experience)."""
        if (os.environ.get("TRLX_AMD_FUSED_LM_LOGPROBS") != "0"
                and h.is_cuda and h.dtype == torch.bfloat16 and lm_head.bias is None
                and h.shape[-1] % 32 == 0):
            B, T = h.shape[:2]
            return ops.lm_logprobs(h.reshape(-1, h.shape[-1]).contiguous(), lm_head.weight,
                                   labels.reshape(-1)).view(B, T)
        return logprobs_of_labels(lm_head(h), labels)

    # --- training ------------------------------------------------------------

    def forward_backward(self, microbatches, loss_fn):
        """1F1B fwd+bwd over the PP group.  ``loss_fn(h_last, mb)`` runs on
        the last stage and returns (loss, stats_dict-of-floats).  Returns
        (mean_loss: float, mean_stats: dict) identically on every PP rank."""
        mean_loss = self.runner.forward_backward(microbatches, loss_fn,
                                                 hidden_dtype=self.hidden_dtype)
        payload = [None]
        if self.stage.is_last:
            stats_list = self.runner.last_stats
            keys = stats_list[0].keys() if stats_list else []
            stats = {k: float(sum(float(s[k]) for s in stats_list)) / max(len(stats_list), 1)
                     for k in keys}
            payload = [(float(mean_loss), stats)]
        if dist.is_initialized():
            dist.broadcast_object_list(payload, src=self.runner.pp_ranks[-1],
                                       group=topo.pp_group())
        return payload[0]

    # --- generation ----------------------------------------------------------

    @torch.no_grad()
    def generate(self, input_ids, attention_mask=None, shaping_fn=None, **kwargs):
        """Token-level pipelined decode (reference NeMo generate parity,
        modeling_nemo_ppo.py:1158-1222): prefill fills per-stage KV caches in
        one pipeline pass; each subsequent token makes one pipeline trip and
        the sampled id is broadcast from the last stage.  All PP ranks return
        the same [B, T+new] samples.  ``shaping_fn(logits, hidden, last_tok)``
        runs on the LAST stage before sampling (ILQL logit shaping)."""
        gen = GenerateConfig.from_kwargs(**kwargs)
        st = self.stage
        device = self.device
        input_ids = input_ids.to(device)
        if attention_mask is None:
            attention_mask = torch.ones_like(input_ids)
        attention_mask = attention_mask.to(device)
        B, T = input_ids.shape
        was_training = self.training
        self.eval()
        try:
            key_starts = (T - attention_mask.sum(-1)).to(torch.int32)
            kv = st.new_kv_cache(B, T + gen.max_new_tokens, device=device)
            hdt = self.hidden_dtype
            pad_id = gen.pad_token_id
            if pad_id is None:
                pad_id = gen.eos_token_id if gen.eos_token_id is not None else 0
            eos_id = gen.eos_token_id
            seed = gen.seed
            if seed is None and st.is_last:
                seed = int(torch.randint(0, 2**31 - 1, (1,)).item())

            ctx = st.make_context(input_ids, attention_mask)
            h = self.runner.forward_inference(input_ids, attention_mask, kv_cache=kv,
                                              ctx=ctx, hidden_dtype=hdt)
            src = self.runner.pp_ranks[-1]
            group = topo.pp_group()
            finished = torch.zeros(B, dtype=torch.bool, device=device)
            generated = []
            last_tok = input_ids[:, -1]
            for step in range(gen.max_new_tokens):
                if st.is_last:
                    logits = st.project(h[:, -1:, :])[:, 0].float()
                    if shaping_fn is not None:
                        logits = shaping_fn(logits, h[:, -1], last_tok)
                    if eos_id is not None and step < gen.min_new_tokens:
                        logits[:, eos_id] = float("-inf")
                    if gen.do_sample:
                        tok = ops.sample_token(logits, gen.temperature, gen.top_k,
                                               gen.top_p, seed=seed, offset=step)
                    else:
                        tok = logits.argmax(dim=-1)
                    tok = torch.where(finished, torch.full_like(tok, pad_id), tok)
                else:
                    tok = torch.empty(B, dtype=torch.long, device=device)
                if dist.is_initialized():
                    dist.broadcast(tok, src=src, group=group)
                generated.append(tok)
                last_tok = tok
                if eos_id is not None:
                    finished = finished | (tok == eos_id)
                    if bool(finished.all()):
                        break
                if step == gen.max_new_tokens - 1:
                    break
                start_pos = T + step
                position_ids = (start_pos - key_starts).to(torch.int32).unsqueeze(1)
                seq_lens = torch.full((B,), start_pos + 1, dtype=torch.int32, device=device)
                step_ctx = AttentionContext(position_ids=position_ids, key_starts=key_starts,
                                            start_pos=start_pos, seq_lens=seq_lens)
                h = self.runner.forward_inference(tok.unsqueeze(1), None, kv_cache=kv,
                                                  ctx=step_ctx, hidden_dtype=hdt)
            return torch.cat([input_ids, torch.stack(generated, dim=1)], dim=1)
        finally:
            if was_training:
                self.train()

    def generate_eval(self, input_ids, attention_mask=None, **kwargs):
        return self.generate(input_ids, attention_mask, **kwargs)

    # --- persistence ---------------------------------------------------------

    def save_pretrained(self, save_directory: str, **kwargs):
        """TP×PP sharded checkpoint: mp_rank_{tp:02d}_{pp:03d}/model_weights.pt
        per model-parallel rank + a config.json carrying the topology.  Every
        model-parallel rank of dp replica 0 must call this (the trainer
        does)."""
        os.makedirs(save_directory, exist_ok=True)
        shard_dir = os.path.join(
            save_directory, f"mp_rank_{topo.tp_rank():02d}_{topo.pp_rank():03d}")
        os.makedirs(shard_dir, exist_ok=True)
        sd = {k: v.cpu() for k, v in self.state_dict().items()
              if "rope_cos" not in k and "rope_sin" not in k}
        torch.save(sd, os.path.join(shard_dir, "model_weights.pt"))
        if topo.tp_rank() == 0 and topo.pp_rank() == 0:
            from .nn.convert import config_to_hf

            with open(os.path.join(save_directory, "config.json"), "w") as f:
                json.dump({**config_to_hf(self.config),
                           "trlx_amd_tp_size": topo.tp_size(),
                           "trlx_amd_pp_size": topo.pp_size(),
                           "trlx_amd_num_layers_unfrozen": self.num_layers_unfrozen},
                          f, indent=2)

    @classmethod
    def from_pretrained(cls, directory: str, **kwargs):
        """Reload a TP×PP shard checkpoint at the SAME topology."""
        with open(os.path.join(directory, "config.json")) as f:
            hf_cfg = json.load(f)
        from .nn.convert import config_from_hf

        cfg = config_from_hf(hf_cfg)
        saved_tp = int(hf_cfg.get("trlx_amd_tp_size", 1))
        saved_pp = int(hf_cfg.get("trlx_amd_pp_size", 1))
        if saved_tp != topo.tp_size() or saved_pp != topo.pp_size():
            raise ValueError(
                f"checkpoint topology tp={saved_tp},pp={saved_pp} != live "
                f"tp={topo.tp_size()},pp={topo.pp_size()}; use "
                f"merge_pp_checkpoint() to reshard")
        kwargs.setdefault("num_layers_unfrozen",
                          int(hf_cfg.get("trlx_amd_num_layers_unfrozen", -1)))
        model = cls(cfg, **kwargs)
        shard = torch.load(
            os.path.join(directory, f"mp_rank_{topo.tp_rank():02d}_{topo.pp_rank():03d}",
                         "model_weights.pt"),
            map_location="cpu", weights_only=True)
        model.load_state_dict(shard, strict=False)
        return model


def merge_pp_checkpoint(directory: str) -> Dict[str, torch.Tensor]:
    """Merge a TP×PP shard checkpoint into ONE full-model wrapper state dict
    (``base_model.*`` + ``v_head.*`` + ``frozen_head.*`` keys) loadable by the
    single-process AutoModelForCausalLMWithHydraValueHead — the PP analogue of
    merge-on-load for TP shards (reference PP resharding,
    modeling_nemo_ppo.py:321-352, inverted)."""
    with open(os.path.join(directory, "config.json")) as f:
        hf_cfg = json.load(f)
    from .nn.convert import config_from_hf

    cfg = config_from_hf(hf_cfg)
    saved_tp = int(hf_cfg.get("trlx_amd_tp_size", 1))
    saved_pp = int(hf_cfg.get("trlx_amd_pp_size", 1))
    from ..parallel.pp import split_layers

    # first: per-TP-rank full dicts assembled from the PP shards
    per_tp = []
    for t in range(saved_tp):
        full = {}
        for p in range(saved_pp):
            shard = torch.load(
                os.path.join(directory, f"mp_rank_{t:02d}_{p:03d}", "model_weights.pt"),
                map_location="cpu", weights_only=True)
            lo, _hi = split_layers(cfg.num_layers, saved_pp, p)
            for k, v in shard.items():
                if k.startswith("stage.layers."):
                    rest = k[len("stage.layers."):]
                    idx, tail = rest.split(".", 1)
                    full[f"base_model.layers.{int(idx) + lo}.{tail}"] = v
                elif k.startswith("stage."):
                    full["base_model." + k[len("stage."):]] = v
                else:
                    full[k] = v  # v_head. / frozen_head.
        per_tp.append(full)
    if saved_tp == 1:
        return per_tp[0]
    from ..parallel.tp import merge_state_dicts_tp

    # merge_state_dicts_tp pattern-matches on submodule names, so the
    # wrapper-prefixed frozen_head.blocks.* keys merge like base layers
    return merge_state_dicts_tp(per_tp, cfg, saved_tp)


class PipelinedILQLModel(PipelinedPPOModel):
    """PP-sharded ILQL model: the Q/V/target-Q heads live on the LAST stage
    (reference NeMo ILQLGPT parity, modeling_nemo_ilql.py:255-785 — heads
    only on the post_process stage, shaped generation inside the pipeline
    inference fn)."""

    def __init__(self, config: TransformerConfig, two_qs: bool = True, alpha: float = 0.99,
                 **kwargs):
        kwargs.pop("with_value_head", None)
        kwargs.setdefault("num_layers_unfrozen", -1)
        super().__init__(config, with_value_head=False, **kwargs)
        self.two_qs = two_qs
        self.alpha = alpha
        self.ilql_heads = None
        if self.stage.is_last:
            from .modeling_ilql import ILQLHeads

            self.ilql_heads = ILQLHeads(config.hidden_size, config.vocab_size, two_qs, alpha)

    def sync_target_q_heads(self):
        if self.ilql_heads is not None:
            self.ilql_heads.sync_target_q_heads()

    @torch.no_grad()
    def generate(self, input_ids, attention_mask=None, beta: float = 1.0,
                 max_new_tokens: int = 32, max_length: int = 1024, temperature: float = 1.0,
                 top_k: int = 20, logit_mask=None, pad_token_id=None, eos_token_id=None,
                 **kwargs):
        """Shaped pipelined generation: log pi + beta*(minQ - V) computed on
        the last stage's hidden (reference modeling_nemo_ilql.py:685-739 —
        the shaping lives inside the pipeline inference fn)."""
        import torch.nn.functional as F

        if attention_mask is None and pad_token_id is not None:
            attention_mask = input_ids.not_equal(pad_token_id).long()

        shaping = None
        if self.stage.is_last:
            def shaping(logits, hidden, last_tokens):
                hs = hidden.unsqueeze(1)
                qs, target_qs, vs = self.ilql_heads(hs)
                if self.two_qs:
                    qv = torch.minimum(target_qs[0][:, -1, :], target_qs[1][:, -1, :])
                else:
                    qv = target_qs[0][:, -1, :]
                v = vs[:, -1, :]
                if logit_mask is not None:
                    m = logit_mask[last_tokens.to(logit_mask.device)]
                    logits = logits.masked_fill(m.to(logits.device), float("-inf"))
                adv = (qv - v).to(logits.dtype)
                return F.log_softmax(logits, -1) + beta * adv

        max_new_tokens = min(max_new_tokens, max_length - input_ids.shape[1])
        return super().generate(
            input_ids, attention_mask, shaping_fn=shaping,
            max_new_tokens=max_new_tokens, do_sample=temperature > 0,
            temperature=temperature if temperature > 0 else 1.0, top_k=top_k,
            eos_token_id=eos_token_id,
            pad_token_id=eos_token_id if eos_token_id is not None else pad_token_id,
        )

    def generate_eval(self, *args, **kwargs):
        return self.generate(*args, **kwargs)

    @classmethod
    def from_any(cls, path_or_config, two_qs=True, alpha=0.99, **kwargs):
        heads = None
        if isinstance(path_or_config, TransformerConfig):
            cfg = path_or_config
            full_sd = None
        elif isinstance(path_or_config, str) and os.path.isdir(path_or_config):
            from .nn.convert import load_hf_dir

            cfg, full_sd = load_hf_dir(path_or_config)
        elif isinstance(path_or_config, str) and path_or_config in PRESETS:
            cfg = preset(path_or_config)
            full_sd = None
        else:
            raise OSError(f"cannot build a pipelined ILQL model from {path_or_config!r}")
        model = cls(cfg, two_qs=two_qs, alpha=alpha, **kwargs)
        if full_sd is not None:
            model.load_full_base_state_dict(full_sd)
        return model
