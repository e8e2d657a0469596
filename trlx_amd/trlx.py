"""Public API: ``trlx_amd.train()``.

Parity target: reference trlx/trlx.py — the single entry point dispatching on
(reward_fn | samples+rewards) to online (PPO/RFT) or offline (ILQL/SFT)
training, building prompt/eval pipelines, optional resume, then learn().
"""

import os
import warnings
from typing import Callable, Dict, Iterable, List, Optional, Tuple, Union

from .data.configs import TRLConfig
from .data.default_configs import (
    default_ilql_config,
    default_ppo_config,
    default_sft_config,
)
from .utils import set_seed
from .utils.loading import get_pipeline, get_trainer
from .utils import logging

logger = logging.get_logger(__name__)


def train(  # noqa: C901
    model_path: Optional[str] = None,
    reward_fn: Optional[Callable[[List[str], List[str], List[str]], List[float]]] = None,
    dataset: Optional[Iterable[Tuple[str, float]]] = None,
    samples: Optional[List[str]] = None,
    rewards: Optional[List[float]] = None,
    prompts: Optional[Union[List[str], List[Dict]]] = None,
    eval_prompts: Optional[Union[List[str], List[Dict]]] = None,
    metric_fn: Optional[Callable[[List[str], List[str], List[str]], Dict[str, List[float]]]] = None,
    config: Optional[TRLConfig] = None,
    logit_mask=None,
    stop_sequences: Optional[List[str]] = [],
):
    """Run training on MI355X (semantics of reference trlx/trlx.py:15-142).

    Dispatch: ``reward_fn`` -> online RL (PPO by default); ``samples`` +
    ``rewards`` -> offline RL (ILQL); ``samples`` alone -> SFT.
    """
    if config is None:
        warnings.warn(
            "Passing the `config` argument implicitly is depreciated, use or"
            "adapt some from `trlx_amd/data/default_configs.py` instead"
        )
        if reward_fn:
            config = default_ppo_config()
        elif rewards:
            config = default_ilql_config()
        else:
            config = default_sft_config()

    set_seed(config.train.seed)

    if dataset:
        warnings.warn("the `dataset` argument is being depreciated, split it into `samples` and `rewards` instead")
        samples, rewards = dataset

    if model_path:
        config.model.model_path = model_path

    trainer = get_trainer(config.train.trainer)(
        config=config,
        reward_fn=reward_fn,
        metric_fn=metric_fn,
        stop_sequences=stop_sequences,
        logit_mask=logit_mask,
        **config.train.trainer_kwargs,
    )

    batch_size = config.train.batch_size * int(os.environ.get("WORLD_SIZE", 1))
    max_prompt_length = config.train.seq_length - config.method.gen_kwargs.get("max_new_tokens", 0)

    # online training (PPO/RFT): collect rollouts with a reward function
    if reward_fn:
        prompts = prompts or [trainer.tokenizer.bos_token] * batch_size
        if eval_prompts is None:
            eval_prompts = prompts[:batch_size]
        pipeline = get_pipeline(config.train.pipeline)(
            prompts, max_prompt_length, trainer.tokenizer
        )
        trainer.add_prompt_pipeline(pipeline)
        if eval_prompts is None:
            eval_prompts = prompts[:batch_size]

    # offline training from reward-labeled data (ILQL)
    elif samples is not None and rewards is not None:
        if len(samples) != len(rewards):
            raise ValueError(f"Number of samples {len(samples)} should match the number of rewards {len(rewards)}")
        if eval_prompts is None:
            eval_prompts = [trainer.tokenizer.bos_token] * batch_size
        trainer.make_experience(samples, rewards, config.train.seq_length)

    # supervised training (SFT)
    elif samples is not None:
        if eval_prompts is None:
            eval_prompts = [trainer.tokenizer.bos_token] * batch_size
        trainer.make_experience(samples, config.train.seq_length)

    else:
        raise ValueError("Either `samples` or `reward_fn` should be given for training")

    eval_pipeline = get_pipeline(config.train.pipeline)(
        eval_prompts, max_prompt_length, trainer.tokenizer
    )
    trainer.add_eval_pipeline(eval_pipeline)

    if config.train.resume_from_checkpoint and os.path.exists(config.train.resume_from_checkpoint):
        trainer.load(config.train.resume_from_checkpoint)

    trainer.learn()
    return trainer
