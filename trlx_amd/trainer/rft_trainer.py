"""RFT trainer — rejection-sampling fine-tuning.

Parity target: reference trlx/trainer/accelerate_rft_trainer.py: per growth
step generate ``n_generations_per_prompt`` samples per prompt, score on rank 0
and broadcast, keep generations above a per-prompt score percentile that rises
linearly from start_percentile to end_percentile over ``n_improve_steps``,
deduplicate, and fine-tune on the survivors with plain CE.
"""

import itertools
from collections import defaultdict
from dataclasses import dataclass, field
from typing import Any, Dict

import numpy as np
import torch
import torch.nn.functional as F

from ..data.configs import TRLConfig
from ..data.method_configs import MethodConfig, register_method
from ..parallel import comm
from ..pipeline.offline_pipeline import PromptPipeline
from ..trainer import register_trainer
from ..utils import logging
from .base_trainer import NativeRLTrainer
from .sft_trainer import CausalLMWrapper

logger = logging.get_logger(__name__)


@dataclass
@register_method
class RFTConfig(MethodConfig):
    """RFT config (reference accelerate_rft_trainer.py:18).

    :param start_percentile: starting score-threshold percentile per prompt
    :param end_percentile: final percentile
    :param n_improve_steps: improvement steps per growth step
    :param n_generations_per_prompt: samples per prompt per growth step
    """

    name: str = "RFTConfig"
    gen_kwargs: Dict[str, Any] = field(default_factory=lambda: dict(max_new_tokens=40, top_k=0, top_p=1.0, do_sample=True))
    start_percentile: float = 0.7
    end_percentile: float = 0.95
    n_improve_steps: int = 4
    n_generations_per_prompt: int = 32


@register_trainer
class RFTTrainer(NativeRLTrainer):
    def __init__(self, config: TRLConfig, **kwargs):
        super().__init__(config, **kwargs)
        self.generate_kwargs = dict(
            config.method.gen_kwargs,
            eos_token_id=self.tokenizer.eos_token_id,
            pad_token_id=self.tokenizer.pad_token_id,
        )
        self.generate_experience_kwargs = None

    def get_arch(self, config: TRLConfig):
        path = config.model.model_path
        if isinstance(path, str) and config.model.model_extra_configs.get("config"):
            from ..models.nn.config import TransformerConfig

            return CausalLMWrapper.from_config(
                TransformerConfig.from_dict(config.model.model_extra_configs["config"])
            )
        return CausalLMWrapper.from_pretrained(path)

    def loss(self, batch):
        input_ids = batch["input_ids"].to(self.device)
        attention_mask = batch["attention_mask"].to(self.device)
        out = self.model(input_ids, attention_mask=attention_mask)
        logits = out.logits[:, :-1, :].float()
        labels = input_ids[:, 1:].clone()
        labels[attention_mask[:, 1:] == 0] = -100
        loss = F.cross_entropy(logits.reshape(-1, logits.shape[-1]), labels.reshape(-1),
                               ignore_index=-100)
        return loss, {"loss": loss.item()}

    def create_train_dataloader(self):
        return self.store.create_loader(self.config.train.batch_size)

    def prepare_learning(self):
        self.epoch_count = 0
        self.iter_count = 0
        self.n_inner_epochs = 1
        # variable number of surviving samples per improvement step
        self.total_steps = self.config.train.total_steps
        self.generations_per_prompt = defaultdict(list)
        self.eval_dataloader = self.eval_pipeline.create_loader(self.config.train.batch_size)
        self.make_experience()

    def add_prompt_pipeline(self, pipeline: PromptPipeline):
        self.prompt_dataloader = pipeline.create_loader(self.config.train.batch_size)

    def post_epoch_callback(self):
        self.make_experience()
        self.epoch_count += 1

    def make_experience(self):  # noqa: C901
        """Generate-score-filter (reference accelerate_rft_trainer.py:117-197)."""
        if self.epoch_count % self.config.method.n_improve_steps == 0:
            generations = []
            for batch in self.prompt_dataloader:
                for _ in range(self.config.method.n_generations_per_prompt):
                    samples = self.generate(batch["input_ids"], batch["attention_mask"])
                    _, str_prompts, str_outputs = self.decode(
                        batch["input_ids"], samples, append_eos_token=True
                    )
                    generations.extend(
                        {"prompt": p, "output": o} for p, o in zip(str_prompts, str_outputs)
                    )

            if torch.distributed.is_initialized():
                all_gens = comm.gather_object(generations)
                generations = list(itertools.chain(*all_gens))

            if comm.is_main_process():
                all_scores = self.reward_fn(
                    samples=[x["prompt"] + x["output"] for x in generations],
                    prompts=[x["prompt"] for x in generations],
                    outputs=[x["output"] for x in generations],
                )
                all_scores = torch.tensor(all_scores, device=self.device, dtype=torch.float)
            else:
                all_scores = torch.zeros(len(generations), device=self.device)
            if torch.distributed.is_initialized():
                torch.distributed.broadcast(all_scores, src=0)
            scores = all_scores

            for g, s in zip(generations, scores):
                self.generations_per_prompt[g["prompt"]].append(
                    {"output": g["output"], "score": s.item()}
                )

        scores = [[x["score"] for x in self.generations_per_prompt[p]] for p in self.generations_per_prompt]

        percentile_delta = (
            self.config.method.end_percentile - self.config.method.start_percentile
        ) / self.config.method.n_improve_steps
        percentile = self.config.method.start_percentile + percentile_delta * (
            self.epoch_count % self.config.method.n_improve_steps
        )
        thresholds = np.array([np.quantile(np.array(s), percentile) for s in scores])
        # quantized rewards: exclude min values, keep max values
        thresholds = np.clip(thresholds, thresholds.min() + 1e-3, thresholds.max() - 1e-3)

        samples_selected = []
        for prompt, threshold in zip(self.generations_per_prompt, thresholds):
            for x in self.generations_per_prompt[prompt]:
                if x["score"] >= threshold:
                    samples_selected.append((prompt, x["output"]))
        samples_selected = list({tuple(x) for x in samples_selected})

        self.tracker.log(
            {
                "scores_mean": float(np.mean(np.hstack(scores))),
                "len_samples_selected": len(samples_selected),
            },
            step=self.iter_count,
        )

        if len(samples_selected):
            joined = [p + o for p, o in samples_selected]
            self.store = PromptPipeline(
                joined, max_prompt_length=2048, tokenizer=self.tokenizer, add_special_tokens=True
            )
