"""ILQL trainer (offline RL on reward-labeled samples).

Parity target: reference trlx/trainer/accelerate_ilql_trainer.py — the
module-level make_experience (30-100: tokenize dialogues, build
actions/states/dones index tensors, return-normalize rewards with the
terminal reward on the last action) and the trainer (target-Q sync every
``steps_for_target_q_sync``, loss via ILQLConfig).
"""

from typing import Union

import numpy as np
import torch

from ..data.configs import TRLConfig
from ..data.ilql_types import ILQLBatch
from ..models.modeling_ilql import AutoModelForCausalLMWithILQLHeads, ILQLConfig
from ..pipeline.offline_pipeline import ILQLRolloutStorage, tokenize_dialogue
from ..trainer import register_trainer
from ..utils import to_device
from ..utils import logging
from .base_trainer import NativeRLTrainer

logger = logging.get_logger(__name__)


def make_experience(samples, rewards, tokenizer=None, max_length=2048, verbose=True) -> ILQLRolloutStorage:
    """Tokenize reward-labeled samples into ILQL tensors
    (reference accelerate_ilql_trainer.py:30-100).

    For each dialogue: ``actions_ixs`` are positions whose NEXT token is a
    model output; ``states_ixs`` appends the final position; ``dones`` is 1
    for every state but the last; returns are normalized across the dataset
    and placed as the terminal reward of each sample.
    """
    if verbose:
        logger.info("Collecting rollouts")
    if tokenizer is not None:
        samples = [tokenize_dialogue(s, tokenizer, max_length) for s in samples]

    all_input_ids = []
    all_actions_ixs = []
    all_states_ixs = []
    all_dones = []
    for sample in samples:
        length = 0
        all_input_ids.append(torch.tensor(sum((s.tokens for s in sample), ())))
        actions_ixs = []
        for dm in sample:
            if dm.is_output:
                actions_ixs.append(torch.arange(length - 1, length + len(dm.tokens) - 1))
            length += len(dm.tokens)
        states_ixs = torch.hstack((*actions_ixs, torch.tensor(length - 1)))
        all_dones.append(torch.tensor([1] * (len(states_ixs) - 1) + [0], dtype=int))
        all_actions_ixs.append(torch.hstack(actions_ixs))
        all_states_ixs.append(states_ixs)

    sample_lengths = np.array(list(map(len, all_input_ids)))
    output_lengths = np.array(list(map(len, all_actions_ixs)))
    prompt_lengths = sample_lengths - output_lengths
    returns = torch.tensor(rewards, dtype=float)

    if verbose:
        logger.info(
            "Experience stats: prompt length %.2f ∈ [%d, %d], output length %.2f ∈ [%d, %d]",
            prompt_lengths.mean(), prompt_lengths.min(), prompt_lengths.max(),
            output_lengths.mean(), output_lengths.min(), output_lengths.max(),
            ranks=[0],
        )

    returns = returns - returns.mean()
    std_returns = returns.std()
    if not torch.isnan(std_returns):
        returns = returns / (std_returns + torch.finfo(returns.dtype).eps)
    rewards = [torch.zeros(len(x)) for x in all_actions_ixs]
    for rs, ret in zip(rewards, returns):
        rs[-1] = ret

    attention_mask = [torch.ones(len(x), dtype=int) for x in all_input_ids]

    return ILQLRolloutStorage(
        all_input_ids, attention_mask, rewards, all_states_ixs, all_actions_ixs, all_dones
    )


@register_trainer
class ILQLTrainer(NativeRLTrainer):
    def __init__(self, config: TRLConfig, **kwargs):
        super().__init__(config, **kwargs)
        if not isinstance(config.method, ILQLConfig):
            raise ValueError("config.method must be ILQLConfig")
        self.ilql: ILQLConfig = config.method
        self.generate_kwargs = dict(
            config.method.gen_kwargs,
            max_length=self.max_length,
            logit_mask=self.logit_mask,
            eos_token_id=self.tokenizer.eos_token_id if self.tokenizer else 0,
            pad_token_id=self.tokenizer.pad_token_id if self.tokenizer else 0,
        )

    def get_arch(self, config: TRLConfig):
        path = config.model.model_path
        kwargs = dict(two_qs=config.method.two_qs, alpha=config.method.alpha,
                      peft_config=config.model.peft_config)
        if config.train.pipeline_parallel_size > 1:
            if config.model.model_arch_type == "seq2seq":
                raise NotImplementedError("seq2seq + pipeline parallelism is not supported")
            from ..models.modeling_pp import PipelinedILQLModel
            from ..models.nn.config import TransformerConfig

            kwargs.pop("peft_config")
            if isinstance(path, str) and config.model.model_extra_configs.get("config"):
                path = TransformerConfig.from_dict(config.model.model_extra_configs["config"])
            return PipelinedILQLModel.from_any(path, **kwargs)
        if config.model.model_arch_type == "seq2seq":
            from ..models.modeling_seq2seq import AutoModelForSeq2SeqLMWithILQLHeads
            from ..models.nn.seq2seq import Seq2SeqConfig

            if isinstance(path, str) and config.model.model_extra_configs.get("config"):
                return AutoModelForSeq2SeqLMWithILQLHeads.from_config(
                    Seq2SeqConfig.from_dict(config.model.model_extra_configs["config"]), **kwargs)
            return AutoModelForSeq2SeqLMWithILQLHeads.from_pretrained(path, **kwargs)
        if isinstance(path, str) and config.model.model_extra_configs.get("config"):
            from ..models.nn.config import TransformerConfig

            return AutoModelForCausalLMWithILQLHeads.from_config(
                TransformerConfig.from_dict(config.model.model_extra_configs["config"]), **kwargs
            )
        return AutoModelForCausalLMWithILQLHeads.from_pretrained(path, **kwargs)

    def post_backward_callback(self):
        if self.iter_count % self.config.method.steps_for_target_q_sync == 0:
            self.unwrapped_model.sync_target_q_heads()

    def pp_train_minibatch(self, microbatches):
        """ILQL loss under pipeline parallelism: logits + Q/V heads on the
        last stage (reference NeMo ILQLGPT get_forward_output_and_loss_func,
        modeling_nemo_ilql.py:612-683)."""
        model = self.model
        device = self.device
        mbs = []
        for batch in microbatches:
            b = to_device(batch, device)
            mbs.append({"input_ids": b.input_ids, "attention_mask": b.attention_mask,
                        "_batch": b})

        def loss_fn(h, mb):
            b = mb["_batch"]
            logits = model.stage.project(h)
            qs, target_qs, vs = model.ilql_heads(h, states_ixs=b.states_ixs,
                                                 actions_ixs=b.actions_ixs)
            loss, stats = self.ilql.loss((logits, (qs, target_qs, vs)), b)
            return loss, {k: float(v) for k, v in stats.items()}

        return model.forward_backward(mbs, loss_fn)

    def loss(self, batch: ILQLBatch):
        batch = to_device(batch, self.device)
        if self.config.model.model_arch_type == "seq2seq":
            out = self.model(
                input_ids=batch.input_ids,
                attention_mask=batch.attention_mask,
                decoder_input_ids=batch.decoder_input_ids,
                actions_ixs=batch.actions_ixs,
                states_ixs=batch.states_ixs,
            )
        else:
            out = self.model(
                input_ids=batch.input_ids,
                attention_mask=batch.attention_mask,
                actions_ixs=batch.actions_ixs,
                states_ixs=batch.states_ixs,
            )
        return self.ilql.loss((out.logits, (out.qs, out.target_qs, out.vs)), batch)

    def create_train_dataloader(self):
        return self.store.create_loader(self.config.train.batch_size)

    def prepare_learning(self):
        self.train_dataloader = self.create_train_dataloader()
        self.eval_dataloader = self.eval_pipeline.create_loader(self.config.train.batch_size)
        self.n_inner_epochs = 1
        self.total_steps = self.config.train.epochs * len(self.train_dataloader)
        self.total_steps = min(self.total_steps, self.config.train.total_steps)

    def make_experience(self, samples, rewards, max_length=2048):
        """Build the ILQL store (called by trlx.train for offline data)."""
        if self.config.model.model_arch_type == "seq2seq":
            return self.make_experience_seq2seq(samples, rewards, max_length)
        self.store = make_experience(samples, rewards, self.tokenizer, max_length=max_length)

    def make_experience_seq2seq(self, samples, rewards, max_length=2048):
        """(prompt, output) pairs -> encoder input + decoder output tensors
        (reference accelerate_ilql_trainer.py:178-244)."""
        from ..pipeline.offline_pipeline import ILQLSeq2SeqRolloutStorage

        logger.info("Collecting rollouts")
        if self.tokenizer:
            samples = [tokenize_dialogue(s, self.tokenizer, max_length) for s in samples]

        all_input_ids = []
        all_output_ids = []
        all_actions_ixs = []
        all_states_ixs = []
        all_dones = []
        for sample in samples:
            all_input_ids.append(torch.tensor(sample[0].tokens))
            all_output_ids.append(torch.tensor(sample[1].tokens))
            actions_ixs = []
            length = 0
            for phrase in sample:
                if phrase.is_output:
                    length = len(phrase.tokens)
                    actions_ixs.append(torch.arange(0, length - 1))
            states_ixs = torch.hstack((*actions_ixs, torch.tensor(length - 1)))
            all_dones.append(torch.tensor([1] * (len(states_ixs) - 1) + [0], dtype=int))
            all_actions_ixs.append(torch.hstack(actions_ixs))
            all_states_ixs.append(states_ixs)

        returns = torch.tensor(rewards, dtype=float)
        returns = (returns - returns.mean()) / (returns.std() + torch.finfo(returns.dtype).eps)
        rewards = [torch.zeros(len(x)) for x in all_actions_ixs]
        for rs, ret in zip(rewards, returns):
            rs[-1] = ret

        attention_mask = [torch.ones(len(x), dtype=int) for x in all_input_ids]
        self.store = ILQLSeq2SeqRolloutStorage(
            all_input_ids, attention_mask, all_output_ids, rewards,
            all_states_ixs, all_actions_ixs, all_dones,
        )
