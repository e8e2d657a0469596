"""PPO trainer.

Parity target: reference trlx/trainer/accelerate_ppo_trainer.py
(AcceleratePPOTrainer): rollout store + KL controller + RunningMoments ctor
(42-106), loss with the left-pad/right-pad index bookkeeping (127-204),
rollout collection make_experience (251-524) with the §2.3 collective
protocol (pad+gather samples, rank-0 reward, scatter scores, AVG-all-reduce
of the KL), and the post-epoch store refresh.

MI355X notes: the policy/value/reference forward during experience runs as a
single trunk pass (hydra ``return_ref_logits=True``); logprob gathers are the
fused HIP kernel; generation rides the KV-cache decode kernel.
"""

import json
import os
import uuid
from time import time
from typing import Any, Dict, List, Tuple

import numpy as np
import torch
import torch.nn.functional as F
from torch.nn.utils.rnn import pad_sequence

from ..data.configs import TRLConfig
from ..data.ppo_types import PPORLBatch, PPORLElement
from .. import ops
from ..models.modeling_ppo import (
    AdaptiveKLController,
    AutoModelForCausalLMWithHydraValueHead,
    AutoModelForCausalLMWithValueHead,
    FixedKLController,
)
from ..parallel import comm, topo
from ..pipeline.offline_pipeline import PromptPipeline
from ..pipeline.ppo_pipeline import PPORolloutStorage
from ..trainer import register_trainer
from ..utils import Clock, infinite_dataloader
from ..utils import logging
from ..utils.modeling import RunningMoments, gather_dict, logprobs_of_labels
from .base_trainer import NativeRLTrainer

logger = logging.get_logger(__name__)


@register_trainer
class PPOTrainer(NativeRLTrainer):
    """PPO RL trainer (also registered under the reference's name)."""

    def __init__(self, config: TRLConfig, **kwargs):
        super().__init__(config, **kwargs)

        self.log_rollouts = config.train.rollout_logging_dir is not None
        if self.log_rollouts:
            self.setup_rollout_logging(config)

        self.store = PPORolloutStorage(self.tokenizer.pad_token_id, self.tokenizer.padding_side)
        self.store.clear_history()

        # separate full reference model only without a hydra branch or peft
        # (reference accelerate_ppo_trainer.py:74-77)
        if getattr(self.model, "is_pipelined", False):
            # PP: the hydra branch on the last stage is the reference; a
            # second pipelined ref model is not supported
            if self.config.model.num_layers_unfrozen <= 0:
                raise NotImplementedError(
                    "PPO under pipeline parallelism requires num_layers_unfrozen > 0 "
                    "(the hydra frozen branch on the last stage is the reference model)"
                )
            self.ref_model = None
        elif (hasattr(self.model, "frozen_head") and self.model.frozen_head is not None) or \
                self.config.model.peft_config is not None:
            self.ref_model = None
        else:
            self.ref_model = self.get_arch(self.config)
            self.ref_model.base_model.load_state_dict(self.model.base_model.state_dict())
            self.ref_model = self.ref_model.to(self.device)
            if self.dtype != torch.float32:
                self.ref_model.cast_compute(self.dtype)
            self.ref_model.eval()

        if config.method.target is not None:
            self.kl_ctl = AdaptiveKLController(config.method.init_kl_coef, config.method.target,
                                               config.method.horizon)
        else:
            self.kl_ctl = FixedKLController(config.method.init_kl_coef)

        self.running_moments = RunningMoments()
        self.ref_mean = config.method.ref_mean
        self.ref_std = config.method.ref_std
        self.mean_kl = 0.0

    def get_arch(self, config: TRLConfig):
        if config.train.pipeline_parallel_size > 1:
            if config.model.model_arch_type == "seq2seq":
                raise NotImplementedError("seq2seq + pipeline parallelism is not supported")
            from ..models.modeling_pp import PipelinedPPOModel
            from ..models.nn.config import TransformerConfig

            path = config.model.model_path
            if (isinstance(path, str) and not os.path.isdir(path)
                    and config.model.model_extra_configs.get("config")):
                path = TransformerConfig.from_dict(config.model.model_extra_configs["config"])
            return PipelinedPPOModel.from_any(
                path, num_layers_unfrozen=config.model.num_layers_unfrozen)
        if config.model.model_arch_type == "seq2seq":
            from ..models.modeling_seq2seq import (
                AutoModelForSeq2SeqLMWithHydraValueHead,
                AutoModelForSeq2SeqLMWithValueHead,
            )
            from ..models.nn.seq2seq import Seq2SeqConfig

            model_cls = AutoModelForSeq2SeqLMWithHydraValueHead
            kwargs = dict(num_layers_unfrozen=config.model.num_layers_unfrozen,
                          peft_config=config.model.peft_config)
            if config.model.num_layers_unfrozen == -1:
                model_cls = AutoModelForSeq2SeqLMWithValueHead
                kwargs.pop("num_layers_unfrozen")
            path = config.model.model_path
            if isinstance(path, str) and not os.path.isdir(path) and config.model.model_extra_configs.get("config"):
                return model_cls.from_config(
                    Seq2SeqConfig.from_dict(config.model.model_extra_configs["config"]), **kwargs)
            return model_cls.from_pretrained(path, **kwargs)
        model_cls = AutoModelForCausalLMWithHydraValueHead
        kwargs = dict(
            num_layers_unfrozen=config.model.num_layers_unfrozen,
            peft_config=config.model.peft_config,
        )
        if config.model.num_layers_unfrozen == -1:
            model_cls = AutoModelForCausalLMWithValueHead
            kwargs.pop("num_layers_unfrozen")
        path = config.model.model_path
        if isinstance(path, str) and not os.path.isdir(path) and hasattr(config.model, "model_extra_configs") and config.model.model_extra_configs.get("config"):
            from ..models.nn.config import TransformerConfig

            return model_cls.from_config(
                TransformerConfig.from_dict(config.model.model_extra_configs["config"]), **kwargs
            )
        return model_cls.from_pretrained(path, **kwargs)

    # --- loss ------------------------------------------------------------------

    def loss(self, batch: PPORLBatch) -> Tuple[torch.Tensor, Dict[str, Any]]:
        """PPO loss on stored rollouts (reference accelerate_ppo_trainer.py:127-204
        — the start/end index semantics are preserved exactly)."""
        query_tensors = batch.query_tensors.to(self.device)
        response_tensors = batch.response_tensors.to(self.device)
        old_logprobs = batch.logprobs.to(self.device)
        old_values = batch.values.to(self.device)
        old_rewards = batch.rewards.to(self.device)
        response_length = old_rewards.shape[1]

        advantages, returns = self.config.method.get_advantages_and_returns(
            old_values, old_rewards, response_length
        )

        if self.config.model.model_arch_type == "seq2seq":
            # encoder gets the query, decoder the response
            # (reference accelerate_ppo_trainer.py:146-174)
            input_ids = query_tensors
            decoder_input_ids = response_tensors
            attention_mask = input_ids.ne(self.tokenizer.pad_token_id).long()
            decoder_attention_mask = decoder_input_ids.ne(self.tokenizer.pad_token_id).long()
            decoder_attention_mask[:, 0] = 1
            outputs = self.model(
                input_ids=input_ids, attention_mask=attention_mask,
                decoder_input_ids=decoder_input_ids,
                decoder_attention_mask=decoder_attention_mask,
            )
            logprobs = logprobs_of_labels(outputs.logits[:, :-1, :], decoder_input_ids[:, 1:])
            mask = decoder_input_ids.ne(self.tokenizer.pad_token_id).long()
            start, end = 0, response_length
            logprobs, values_pred, mask = (
                logprobs[:, start:end],
                outputs.values[:, start:end],
                mask[:, start + 1 : end + 1],
            )
            return self.config.method.loss(
                logprobs=logprobs, values=values_pred, old_logprobs=old_logprobs,
                old_values=old_values, advantages=advantages, returns=returns, mask=mask,
            )

        tokens = torch.cat((query_tensors, response_tensors), dim=1)
        attention_mask = tokens.not_equal(self.tokenizer.pad_token_id).long().to(tokens.device)
        # lm_head/v_head/logprobs run only on the response positions
        # [start, end) — identical math to the reference's full-width compute
        # + slice (accelerate_ppo_trainer.py:178-192), ~2.5x less head work
        start = query_tensors.shape[1] - 1
        end = start + response_length
        labels_sl = tokens[:, start + 1 : end + 1]
        lm = getattr(self.model.base_model, "lm_head", None)
        if (os.environ.get("TRLX_AMD_FUSED_CE") == "1"
                and lm is not None and lm.bias is None and tokens.is_cuda
                and lm.weight.dtype == torch.bfloat16 and lm.weight.shape[1] % 64 == 0):
            # OPT-IN fused train path: logprobs straight from hidden states;
            # the [N, V] logits exist only as the bf16 dlogits of the
            # backward.  Numerically verified, but the backward RECOMPUTES
            # the logits GEMM where the stock path just re-reads the saved
            # bf16 logits (264 MB round trip ~= 42 us vs ~180 us recompute at
            # N=1312): same-box A/B measured 715 vs 730 samples/s, so the
            # stock path stays the default at GPT-2 scale.  The fused path
            # wins when activation memory is the constraint (logits for a
            # [B, T, 50k+] slice can dominate at long response lengths).
            outputs = self.model(tokens, attention_mask, logits_slice=(start, end),
                                 return_logits=False)
            h = outputs.last_hidden_state[:, start:end].contiguous()
            logprobs = ops.lm_logprobs_train(
                h.reshape(-1, h.shape[-1]), lm.weight, labels_sl.reshape(-1)
            ).view(labels_sl.shape)
        else:
            outputs = self.model(tokens, attention_mask, logits_slice=(start, end))
            logprobs = logprobs_of_labels(outputs.logits, labels_sl)
        values_pred = outputs.values
        mask = attention_mask[:, start + 1 : end + 1]

        loss, stats = self.config.method.loss(
            logprobs=logprobs,
            values=values_pred,
            old_logprobs=old_logprobs,
            old_values=old_values,
            advantages=advantages,
            returns=returns,
            mask=mask,
        )
        return loss, stats

    def pp_train_minibatch(self, microbatches):
        """PPO loss under pipeline parallelism: 1F1B microbatched fwd+bwd with
        the loss evaluated on the LAST stage from its post-norm hidden states
        (reference nemo training_step + get_forward_output_and_loss_func,
        modeling_nemo_ppo.py:652-786, 945-1026).  Returns (loss, stats) —
        host floats, identical on every PP rank."""
        method = self.config.method
        model = self.model
        pad_id = self.tokenizer.pad_token_id
        device = self.device

        mbs = []
        for batch in microbatches:
            tokens = torch.cat((batch.query_tensors.to(device),
                                batch.response_tensors.to(device)), dim=1)
            mask = tokens.not_equal(pad_id).long()
            mbs.append({"input_ids": tokens, "attention_mask": mask, "_batch": batch})

        def loss_fn(h, mb):
            batch = mb["_batch"]
            old_logprobs = batch.logprobs.to(device)
            old_values = batch.values.to(device)
            old_rewards = batch.rewards.to(device)
            response_length = old_rewards.shape[1]
            advantages, returns = method.get_advantages_and_returns(
                old_values, old_rewards, response_length)
            start = batch.query_tensors.shape[1] - 1
            end = start + response_length
            hs = h[:, start:end].contiguous()
            labels = mb["input_ids"][:, start + 1 : end + 1]
            logprobs = logprobs_of_labels(model.stage.project(hs), labels)
            values_pred = model.v_head(hs.to(model.v_head[0].weight.dtype)).squeeze(-1).float()
            mask_sl = mb["attention_mask"][:, start + 1 : end + 1]
            loss, stats = method.loss(
                logprobs=logprobs, values=values_pred, old_logprobs=old_logprobs,
                old_values=old_values, advantages=advantages, returns=returns, mask=mask_sl,
            )
            # host floats: the (loss, stats) pair is object-broadcast to the
            # other PP stages after the pipeline drains
            return loss, {k: float(v) for k, v in stats.items()}

        return model.forward_backward(mbs, loss_fn)

    # --- plumbing ----------------------------------------------------------------

    def setup_rollout_logging(self, config):
        exists = os.path.exists(config.train.rollout_logging_dir)
        isdir = os.path.isdir(config.train.rollout_logging_dir)
        assert exists and isdir
        self.run_id = f"run-{uuid.uuid4()}"
        self.rollout_logging_dir = os.path.join(config.train.rollout_logging_dir, self.run_id)
        os.mkdir(self.rollout_logging_dir)
        with open(os.path.join(self.rollout_logging_dir, "config.json"), "w") as f:
            f.write(json.dumps(config.to_dict(), indent=2))

    def post_epoch_callback(self):
        """Clear the store and collect fresh rollouts — the RL outer loop."""
        if self.log_rollouts:
            self.store.export_history(location=self.rollout_logging_dir)
        self.store.clear_history()
        self.make_experience(self.config.method.num_rollouts, self.iter_count)

    def post_backward_callback(self):
        self.kl_ctl.update(self.mean_kl, n_steps=self.config.train.batch_size)

    def create_train_dataloader(self):
        # deterministic per-inner-epoch shuffle seed, identical across ranks:
        # TP/PP peers must draw identical minibatch orders from their
        # identical stores (each DP rank's store holds different data, so DP
        # decorrelation is unaffected)
        self._loader_seed = getattr(self, "_loader_seed", self.config.train.seed) + 1
        return self.store.create_loader(self.config.train.batch_size, shuffle=True,
                                        seed=self._loader_seed)

    def prepare_learning(self):
        self.eval_dataloader = self.eval_pipeline.create_loader(self.config.method.chunk_size)
        self.make_experience(self.config.method.num_rollouts)
        self.train_dataloader = self.create_train_dataloader()
        self.n_inner_epochs = self.config.method.ppo_epochs
        self.total_steps = self.config.train.epochs * self.n_inner_epochs * len(self.train_dataloader)
        self.total_steps = min(self.total_steps, self.config.train.total_steps)

    def add_prompt_pipeline(self, pipeline: PromptPipeline):
        prompt_dataloader = pipeline.create_loader(self.config.method.chunk_size, shuffle=True)
        self.prompt_iterator = infinite_dataloader(prompt_dataloader)

    # --- experience generation ------------------------------------------------------

    def make_experience(self, num_rollouts: int = 1024, iter_count: int = 0):  # noqa: C901
        """Collect rollouts: generate, score on rank 0, scatter scores, compute
        per-token KL-penalty rewards, push PPORLElements
        (reference accelerate_ppo_trainer.py:251-524)."""
        logger.info("Collecting rollouts")
        clock = Clock()
        ppo_rl_batches: List[Tuple[PPORLBatch, torch.Tensor]] = []
        n_collected = 0
        accumulated_stats: List[Dict] = []
        device = self.device
        tbar = logging.tqdm(total=num_rollouts, disable=not comm.is_main_process(),
                            desc=f"[rollout 0 / {num_rollouts}]")

        while n_collected < num_rollouts:
            stats = {}
            batch = next(self.prompt_iterator)

            rollout_generate_time = time()
            samples = self.generate(batch["input_ids"], batch["attention_mask"])
            stats["time/rollout_generate"] = time() - rollout_generate_time

            prompt_tensors = batch["input_ids"].to(device)
            prompt_sizes = torch.full((len(prompt_tensors),), prompt_tensors.shape[1],
                                      device=device, dtype=torch.long)
            metadata_local = {k: v for k, v in batch.items()
                              if k not in ("input_ids", "attention_mask")}

            if (topo.tp_size() > 1 or topo.pp_size() > 1
                    or getattr(self.config.method, "local_rewards", False)):
                # model-parallel mode (or method.local_rewards): every rank
                # scores its own samples locally (TP/PP peers hold identical
                # rollouts; under DP this requires a stateless reward fn) —
                # the NeMo-path protocol (reference nemo_ppo_trainer.py:195-197)
                l_samples, l_prompts, l_outputs = self.decode(
                    prompt_tensors, samples, append_eos_token=True)
                rollout_score_time = time()
                local_scores = self.reward_fn(
                    samples=l_samples, prompts=l_prompts, outputs=l_outputs,
                    tokenizer=self.tokenizer, **metadata_local,
                )
                local_scores = [torch.tensor(score, dtype=torch.float, device=device).view(-1)
                                for score in local_scores]
                scores = pad_sequence(local_scores, batch_first=True, padding_value=-np.inf)
                stats["time/rollout_score"] = time() - rollout_score_time
            else:
                # accelerate-path protocol: gather everything, score on rank 0,
                # scatter per-rank chunks (reference accelerate_ppo_trainer.py:292-338)
                padded_samples = comm.pad_across_processes(samples, 1, self.tokenizer.eos_token_id)
                padded_prompts = comm.pad_across_processes(prompt_tensors, 1, self.tokenizer.eos_token_id)
                gathered_samples = comm.gather(padded_samples)
                gathered_prompts = comm.gather(padded_prompts)
                gathered_prompt_sizes = comm.gather(prompt_sizes)
                metadata = gather_dict(metadata_local)

                if comm.is_main_process():
                    rollout_decode_time = time()
                    all_str_samples, all_str_prompts, all_str_outputs = self.decode(
                        gathered_prompts, gathered_samples, gathered_prompt_sizes, append_eos_token=True
                    )
                    stats["time/rollout_decode"] = time() - rollout_decode_time
                    rollout_score_time = time()
                    all_scores = self.reward_fn(
                        samples=all_str_samples, prompts=all_str_prompts, outputs=all_str_outputs,
                        tokenizer=self.tokenizer, **metadata,
                    )
                    all_scores = [torch.tensor(score, dtype=torch.float, device=device).view(-1)
                                  for score in all_scores]
                    all_scores = pad_sequence(all_scores, batch_first=True, padding_value=-np.inf)
                    max_len = torch.tensor(all_scores.shape[1], dtype=torch.long, device=device)
                    stats["time/rollout_score"] = time() - rollout_score_time
                    all_scores = list(all_scores.reshape(self.world_size, -1, max_len).unbind())
                else:
                    all_scores = None
                    max_len = torch.tensor(0, dtype=torch.long, device=device)

                if torch.distributed.is_initialized():
                    torch.distributed.broadcast(max_len, 0)
                    scores = torch.empty((len(samples), max_len), device=device)
                    torch.distributed.scatter(scores, all_scores)
                else:
                    scores = all_scores[0].clone().detach()
            scores_mask = scores != -np.inf

            if (topo.tp_size() <= 1 and topo.pp_size() <= 1 and self.world_size == 1
                    and not getattr(self.config.method, "local_rewards", False)):
                # single process: the gathered batch IS the local batch
                str_samples, str_prompts, str_outputs = (
                    all_str_samples, all_str_prompts, all_str_outputs)
            else:
                str_samples, str_prompts, str_outputs = self.decode(prompt_tensors, samples,
                                                                    append_eos_token=True)

            # re-tokenize outputs (stop sequences may have trimmed them)
            retok_time = time()
            outputs = self.tokenizer(str_outputs).input_ids
            if self.config.model.model_arch_type == "seq2seq":
                # decoder sequences start with the decoder-start/pad token
                # (reference accelerate_ppo_trainer.py:352-356)
                dst = getattr(self.model.config, "decoder_start_token_id", self.tokenizer.pad_token_id)
                outputs = [[dst] + o for o in outputs]
            outputs = list(map(torch.LongTensor, outputs))
            maxsize = max(max(map(len, outputs)), 1)
            outputs = [
                F.pad(output, (0, maxsize - len(output)), value=self.tokenizer.pad_token_id)
                for output in outputs
            ]
            sample_outputs = torch.vstack(outputs).to(device)
            stats["time/rollout_retokenize"] = time() - retok_time

            if self.config.method.cliprange_reward:
                scores = torch.clip(scores, -self.config.method.cliprange_reward,
                                    self.config.method.cliprange_reward)

            if self.ref_mean is None:
                self.ref_mean = (scores * scores_mask).sum(dim=1).mean()
                self.ref_std = (scores * scores_mask).sum(dim=1).std()
            all_scores_mean, all_scores_std = self.running_moments.update(
                torch.sum(scores * scores_mask, dim=1))
            stats["rollout_scores/mean"] = float(all_scores_mean)
            stats["rollout_scores/std"] = float(all_scores_std)
            stats["rollout_scores/running_mean"] = float(self.running_moments.mean)
            stats["rollout_scores/running_std"] = float(self.running_moments.std)

            if self.config.method.scale_reward == "running":
                scores /= self.running_moments.std
            elif self.config.method.scale_reward == "ref":
                scores /= self.ref_std

            rollout_fwd_time = time()
            if self.config.model.model_arch_type == "seq2seq":
                # seq2seq experience pass (reference accelerate_ppo_trainer.py:383-414)
                n_samples = samples.shape[0]
                enc_mask = batch["attention_mask"].to(device)
                dec_mask = sample_outputs.ne(self.tokenizer.pad_token_id).long()
                dec_mask[:, 0] = 1
                with torch.no_grad():
                    outputs = self.model(
                        input_ids=prompt_tensors, attention_mask=enc_mask,
                        decoder_input_ids=sample_outputs, decoder_attention_mask=dec_mask,
                        return_ref_logits=True,
                    )
                    logits, values = outputs.logits, outputs.values
                    if outputs.ref_logits is not None:
                        ref_logits = outputs.ref_logits
                    elif self.ref_model is not None:
                        ref_logits = self.ref_model(
                            input_ids=prompt_tensors, attention_mask=enc_mask,
                            decoder_input_ids=sample_outputs, decoder_attention_mask=dec_mask,
                        ).logits
                    else:
                        ref_logits = logits
                    logprobs = logprobs_of_labels(logits[:, :-1, :], sample_outputs[:, 1:])
                    ref_logprobs = logprobs_of_labels(ref_logits[:, :-1, :], sample_outputs[:, 1:])

                attn_for_kl = sample_outputs.ne(self.tokenizer.pad_token_id).long().to(device)
                start = 0
                log_ratio = (logprobs - ref_logprobs) * attn_for_kl[:, :-1]
                kl = log_ratio.exp() - 1 - log_ratio
                mean_kl_per_token = kl.mean()
                mean_kl = kl.sum(1).mean()

                values = values[:, :-1]
                ends_t = start + attn_for_kl[:, start:].sum(1) + 1
                resp_logprobs = logprobs[:, start:]
                resp_values = values[:, start:]
                resp_log_ratio = log_ratio[:, start:]
            else:
                # causal experience pass — one trunk pass for policy logits +
                # values (+ ref logits via hydra).  With the FIXED KL
                # controller the vocab-wide lm_head/logprob math runs only on
                # the response region [start, T-1) (mean_kl is then a logged
                # stat, measured over the response region).  With the ADAPTIVE
                # controller mean_kl drives kl_ctl.update, so the full-sequence
                # extent is computed to match the reference's
                # kl.sum(1).mean() over prompt+response
                # (accelerate_ppo_trainer.py:455-460,506).
                all_tokens = torch.cat((prompt_tensors, sample_outputs), dim=1)
                attention_mask = all_tokens.not_equal(self.tokenizer.pad_token_id).long().to(device)
                n_samples = samples.shape[0]
                start = prompt_tensors.shape[1] - 1
                T_all = all_tokens.shape[1]
                kl_lo = 0 if isinstance(self.kl_ctl, AdaptiveKLController) else start
                labels = all_tokens[:, kl_lo + 1 :]
                with torch.no_grad():
                    if hasattr(self.model, "forward_experience"):
                        # fused hand-MFMA path: hidden -> logprobs directly,
                        # the [B, T, V] logits never materialize
                        logprobs, ref_logprobs, values = self.model.forward_experience(
                            all_tokens, attention_mask, kl_lo, T_all - 1, labels)
                        if ref_logprobs is None:
                            if self.ref_model is not None:
                                ref_logprobs, _, _ = self.ref_model.forward_experience(
                                    all_tokens, attention_mask, kl_lo, T_all - 1, labels)
                            else:
                                # num_layers_unfrozen == -1, no ref: KL vs itself
                                ref_logprobs = logprobs
                    else:
                        outputs = self.model(all_tokens, attention_mask=attention_mask,
                                             return_ref_logits=True,
                                             logits_slice=(kl_lo, T_all - 1))
                        logits, values = outputs.logits, outputs.values
                        if outputs.ref_logits is not None:
                            ref_logits = outputs.ref_logits
                        elif self.ref_model is not None:
                            ref_logits = self.ref_model(all_tokens, attention_mask=attention_mask,
                                                        logits_slice=(kl_lo, T_all - 1)).logits
                        else:
                            ref_logits = logits
                        logprobs = logprobs_of_labels(logits, labels)
                        ref_logprobs = logprobs_of_labels(ref_logits, labels)

                log_ratio = (logprobs - ref_logprobs) * attention_mask[:, kl_lo:-1]
                kl = log_ratio.exp() - 1 - log_ratio
                mean_kl_per_token = kl.mean()
                mean_kl = kl.sum(1).mean()

                off = start - kl_lo  # 0 on the fixed-KL fast path
                ends_t = attention_mask[:, start:].sum(1) + 1
                resp_logprobs = logprobs[:, off:]
                resp_values = values[:, off:]
                resp_log_ratio = log_ratio[:, off:]

            if torch.cuda.is_available():
                torch.cuda.synchronize()
            stats["time/rollout_forward"] = time() - rollout_fwd_time
            # rollout tensors stay DEVICE-resident as whole padded chunk
            # batches — no CPU round trip and no per-element slicing (the
            # collate path cost ~2.6k tiny device copies per cycle,
            # profile r01).  Semantics match the reference's per-element
            # loop (accelerate_ppo_trainer.py:472-500) exactly: valid width
            # per row is ends, terminal score lands on the last valid token.
            R = resp_logprobs.shape[1]
            ends_t = ends_t.clamp(max=R)
            valid = torch.arange(R, device=device).unsqueeze(0) < ends_t.unsqueeze(1)
            rewards_b = (self.kl_ctl.value * -resp_log_ratio).clone()
            if scores.shape[1] == 1:
                # terminal reward at the last (eos) token
                rewards_b.scatter_add_(1, (ends_t - 1).clamp(min=0).unsqueeze(1),
                                       scores[:, :1].to(rewards_b.dtype))
            else:
                # dense per-token rewards
                sc = (scores * scores_mask).to(rewards_b.dtype)
                w = min(sc.shape[1], R)
                rewards_b[:, :w] = rewards_b[:, :w] + sc[:, :w]
            rewards_b = rewards_b * valid
            wmax = int(ends_t.max())
            chunk_batch = PPORLBatch(
                prompt_tensors,
                sample_outputs,
                (resp_logprobs * valid)[:, :wmax],
                (resp_values[:, :R] * valid)[:, :wmax],
                rewards_b[:, :wmax],
            )
            ppo_rl_batches.append((chunk_batch, ends_t.to("cpu", torch.long)))
            rollout_count = n_samples
            n_collected += n_samples

            if torch.distributed.is_initialized():
                comm.all_reduce_mean(mean_kl)

            stats["time/rollout_time"] = clock.tick()
            stats["policy/sqrt_kl"] = torch.sqrt(torch.clamp(mean_kl, min=0)).item()
            stats["policy/kl_per_token"] = torch.sqrt(torch.clamp(mean_kl_per_token, min=0)).item()
            accumulated_stats.append(stats)
            tbar.set_description(f"[rollout {n_collected} / {num_rollouts}]")
            tbar.update(min(rollout_count, num_rollouts))
        tbar.close()

        stats = {k: sum(xs.get(k, 0.0) for xs in accumulated_stats) / len(accumulated_stats)
                 for k in accumulated_stats[-1]}
        stats["kl_ctl_value"] = self.kl_ctl.value
        self.mean_kl = stats["policy/sqrt_kl"] ** 2
        self.last_experience_stats = stats
        self.tracker.log(stats, step=iter_count)

        for chunk_b, lengths in ppo_rl_batches:
            self.store.push_batch(chunk_b, lengths)

    def save_pretrained(self, directory=None, **kwargs):
        """PPO export saves the base model only (reference ppo_trainer:526-553)."""
        super().save_pretrained(directory, **kwargs)
