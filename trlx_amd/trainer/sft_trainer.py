"""Supervised fine-tuning trainer.

Parity target: reference trlx/trainer/accelerate_sft_trainer.py — plain
causal-LM cross-entropy with attention-masked (or -100-masked) labels;
store is PromptPipeline (plain text) or DialogStore (prompt/output pairs).
"""

from dataclasses import dataclass, field
from typing import Any, Dict

import torch
import torch.nn.functional as F

from ..data.configs import TRLConfig
from ..data.method_configs import MethodConfig, register_method
from ..models.modeling_base import PreTrainedModelWrapper
from ..models.nn.generation import generate
from ..pipeline.offline_pipeline import DialogStore, PromptPipeline, tokenize_dialogue
from ..trainer import register_trainer
from .base_trainer import NativeRLTrainer


@dataclass
@register_method
class SFTConfig(MethodConfig):
    """SFT config (reference accelerate_sft_trainer.py:16)."""

    name: str = "SFTConfig"
    gen_kwargs: Dict[str, Any] = field(default_factory=lambda: dict(max_new_tokens=40, top_k=0, top_p=1.0, do_sample=True))


class CausalLMWrapper(PreTrainedModelWrapper):
    """Bare LM wrapper (no heads) with generate."""

    _supported_args = ["peft_config"]

    def __init__(self, base_model, peft_config=None):
        super().__init__(base_model)
        self.peft_config = peft_config

    def forward(self, input_ids, attention_mask=None, position_ids=None, **kwargs):
        return self.base_model(input_ids, attention_mask=attention_mask, position_ids=position_ids)

    def generate(self, input_ids, attention_mask=None, **kwargs):
        return generate(self.base_model, input_ids, attention_mask, **kwargs)


@register_trainer
class SFTTrainer(NativeRLTrainer):
    def __init__(self, config: TRLConfig, **kwargs):
        super().__init__(config, **kwargs)
        self.generate_kwargs = dict(
            config.method.gen_kwargs,
            eos_token_id=self.tokenizer.eos_token_id,
            pad_token_id=self.tokenizer.pad_token_id,
        )

    def get_arch(self, config: TRLConfig):
        path = config.model.model_path
        if config.train.pipeline_parallel_size > 1:
            from ..models.modeling_pp import PipelinedPPOModel
            from ..models.nn.config import TransformerConfig

            if isinstance(path, str) and config.model.model_extra_configs.get("config"):
                path = TransformerConfig.from_dict(config.model.model_extra_configs["config"])
            return PipelinedPPOModel.from_any(
                path, num_layers_unfrozen=config.model.num_layers_unfrozen,
                with_value_head=False)
        if isinstance(path, str) and config.model.model_extra_configs.get("config"):
            from ..models.nn.config import TransformerConfig

            return CausalLMWrapper.from_config(
                TransformerConfig.from_dict(config.model.model_extra_configs["config"]),
                peft_config=config.model.peft_config,
            )
        return CausalLMWrapper.from_pretrained(path, peft_config=config.model.peft_config)

    def pp_train_minibatch(self, microbatches):
        """SFT CE loss under pipeline parallelism (reference NeMo SFT path,
        modeling_nemo_sft.py training_step skeleton)."""
        model = self.model
        device = self.device
        mbs = []
        for batch in microbatches:
            if isinstance(batch, dict):
                ids = batch["input_ids"].to(device)
                mask = batch.get("attention_mask")
                labels = batch.get("labels", ids)
            else:
                ids = batch.input_ids.to(device)
                mask = getattr(batch, "attention_mask", None)
                labels = ids
            mask = mask.to(device) if mask is not None else torch.ones_like(ids)
            mbs.append({"input_ids": ids, "attention_mask": mask,
                        "_labels": labels.to(device)})

        def loss_fn(h, mb):
            logits = model.stage.project(h)[:, :-1, :].float()
            shift_labels = mb["_labels"][:, 1:].clone()
            shift_labels[mb["attention_mask"][:, 1:] == 0] = -100
            loss = F.cross_entropy(
                logits.reshape(-1, logits.shape[-1]), shift_labels.reshape(-1),
                ignore_index=-100,
            )
            return loss, {"loss": float(loss)}

        return model.forward_backward(mbs, loss_fn)

    def loss(self, batch):
        if isinstance(batch, dict):
            input_ids = batch["input_ids"].to(self.device)
            attention_mask = batch.get("attention_mask")
            labels = batch.get("labels", input_ids)
        else:
            input_ids = batch.input_ids.to(self.device)
            attention_mask = getattr(batch, "attention_mask", None)
            labels = input_ids
        if attention_mask is not None:
            attention_mask = attention_mask.to(self.device)
        labels = labels.to(self.device)

        out = self.model(input_ids, attention_mask=attention_mask)
        logits = out.logits[:, :-1, :].float()
        shift_labels = labels[:, 1:].clone()
        if attention_mask is not None:
            shift_labels[attention_mask[:, 1:] == 0] = -100
        loss = F.cross_entropy(
            logits.reshape(-1, logits.shape[-1]), shift_labels.reshape(-1), ignore_index=-100
        )
        return loss, {"loss": loss.item()}

    def prepare_learning(self):
        self.eval_dataloader = self.eval_pipeline.create_loader(self.config.train.batch_size)
        self.train_dataloader = self.store.create_loader(self.config.train.batch_size, shuffle=True)
        self.n_inner_epochs = 1
        self.total_steps = self.config.train.epochs * len(self.train_dataloader)
        self.total_steps = min(self.total_steps, self.config.train.total_steps)

    def create_train_dataloader(self):
        return self.store.create_loader(self.config.train.batch_size, shuffle=True)

    def make_experience(self, samples, seq_length):
        """Build the SFT store from raw strings or (prompt, output) dialogs
        (reference accelerate_sft_trainer.py:92-97)."""
        if isinstance(samples[0], str):
            self.store = PromptPipeline(samples, seq_length, self.tokenizer)
        else:
            dialogs = [tokenize_dialogue(d, self.tokenizer, seq_length) for d in samples]
            self.store = DialogStore(dialogs, self.tokenizer)
