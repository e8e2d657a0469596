"""ILQL on the randomwalks task (parity: reference examples/randomwalks/
ilql_randomwalks.py) — offline RL from reward-labeled walks."""

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
from trlx_amd.models.modeling_ilql import ILQLConfig
from trlx_amd.models.nn.config import TransformerConfig

model_config = TransformerConfig(
    vocab_size=128, hidden_size=144, num_layers=6, num_heads=6, head_dim=24,
    max_position_embeddings=64, arch_name="gpt2",
)

default_config = TRLConfig(
    train=TrainConfig(
        seq_length=11,
        batch_size=100,
        epochs=20,
        total_steps=1000,
        checkpoint_interval=1000,
        eval_interval=16,
        pipeline="PromptPipeline",
        trainer="ILQLTrainer",
    ),
    model=ModelConfig(model_path="randomwalks", num_layers_unfrozen=-1,
                      model_extra_configs={"config": model_config.to_dict()}),
    tokenizer=TokenizerConfig(tokenizer_path="byte", truncation_side="right"),
    optimizer=OptimizerConfig(name="fused_adamw", kwargs=dict(lr=2e-4, betas=(0.9, 0.95), eps=1.0e-8, weight_decay=1.0e-6)),
    scheduler=SchedulerConfig(name="cosine_annealing", kwargs=dict(T_max=1000, eta_min=2e-4)),
    method=ILQLConfig(
        name="ILQLConfig",
        tau=0.8,
        gamma=0.99,
        cql_scale=0.1,
        awac_scale=1,
        alpha=0.1,
        beta=0,
        steps_for_target_q_sync=5,
        two_qs=True,
        gen_kwargs=dict(max_new_tokens=9, top_k=10, beta=[0, 1, 100], temperature=1.0),
    ),
)


def main(hparams={}):
    config = TRLConfig.update(default_config.to_dict(), hparams)
    metric_fn, eval_prompts, walks, _ = generate_random_walks(seed=config.train.seed)
    rewards = metric_fn(walks)["optimality"]
    # split each walk into (starting state, rest of the walk)
    walks = [[walk[:1], walk[1:]] for walk in walks]

    trlx.train(
        samples=walks,
        rewards=rewards,
        eval_prompts=eval_prompts,
        metric_fn=lambda samples, **kwargs: metric_fn(samples),
        config=config,
    )


if __name__ == "__main__":
    hparams = {} if len(sys.argv) == 1 else json.loads(sys.argv[1])
    main(hparams)
