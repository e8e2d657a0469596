"""RFT on the randomwalks task (parity: reference rft_randomwalks.py)."""

import json
import sys

import os as _os
import sys as _sys
_sys.path.insert(0, _os.path.dirname(_os.path.dirname(_os.path.dirname(_os.path.abspath(__file__)))))
import trlx_amd as trlx
from examples.randomwalks import generate_random_walks
from trlx_amd.data.configs import (
    ModelConfig,
    OptimizerConfig,
    SchedulerConfig,
    TokenizerConfig,
    TrainConfig,
    TRLConfig,
)
from trlx_amd.models.nn.config import TransformerConfig
from trlx_amd.trainer.rft_trainer import RFTConfig

model_config = TransformerConfig(
    vocab_size=128, hidden_size=144, num_layers=6, num_heads=6, head_dim=24,
    max_position_embeddings=64, arch_name="gpt2",
)

default_config = TRLConfig(
    train=TrainConfig(
        seq_length=10,
        epochs=100,
        total_steps=1000,
        batch_size=100,
        checkpoint_interval=10000,
        eval_interval=20,
        pipeline="PromptPipeline",
        trainer="RFTTrainer",
    ),
    model=ModelConfig(model_path="randomwalks", num_layers_unfrozen=-1,
                      model_extra_configs={"config": model_config.to_dict()}),
    tokenizer=TokenizerConfig(tokenizer_path="byte", truncation_side="right"),
    optimizer=OptimizerConfig(name="fused_adamw", kwargs=dict(lr=3.0e-4, betas=(0.9, 0.95), eps=1.0e-8, weight_decay=1.0e-6)),
    scheduler=SchedulerConfig(name="cosine_annealing", kwargs=dict(T_max=1000, eta_min=3.0e-4)),
    method=RFTConfig(
        name="RFTConfig",
        n_generations_per_prompt=16,
        start_percentile=0.9,
        end_percentile=0.95,
        n_improve_steps=1,
        gen_kwargs=dict(max_new_tokens=9, top_k=0, top_p=1.0, do_sample=True),
    ),
)


def main(hparams={}):
    config = TRLConfig.update(default_config.to_dict(), hparams)
    metric_fn, prompts, *_ = generate_random_walks(seed=config.train.seed)

    trlx.train(
        reward_fn=lambda samples, **kwargs: metric_fn(samples)["optimality"],
        prompts=prompts,
        eval_prompts=prompts,
        metric_fn=lambda samples, **kwargs: metric_fn(samples),
        config=config,
    )


if __name__ == "__main__":
    hparams = {} if len(sys.argv) == 1 else json.loads(sys.argv[1])
    main(hparams)
