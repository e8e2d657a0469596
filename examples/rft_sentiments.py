"""RFT sentiments (parity: reference examples/rft_sentiments.py) — rejection
sampling fine-tuning against the sentiment reward."""

import json
import sys

import os as _os
import sys as _sys
_sys.path.insert(0, _os.path.dirname(_os.path.dirname(_os.path.abspath(__file__))))
import trlx_amd as trlx
from trlx_amd.data.default_configs import default_sft_config
from trlx_amd.models.nn.config import preset
from trlx_amd.trainer.rft_trainer import RFTConfig

from ppo_sentiments import EVAL_PROMPTS, PROMPTS, sentiment_reward


def main(hparams={}):
    config = default_sft_config()
    config.train.trainer = "RFTTrainer"
    config.model.model_path = "gpt2"
    config.model.model_extra_configs = {"config": preset("gpt2").to_dict()}
    config.tokenizer.tokenizer_path = "byte"
    config.train.seq_length = 128
    config.train.batch_size = 16
    config.method = RFTConfig(
        name="RFTConfig",
        gen_kwargs=dict(max_new_tokens=24, top_k=0, top_p=1.0, do_sample=True),
        n_generations_per_prompt=8,
        start_percentile=0.7,
        end_percentile=0.95,
        n_improve_steps=4,
    )
    config = trlx.TRLConfig.update(config.to_dict(), hparams)

    trlx.train(
        reward_fn=sentiment_reward,
        prompts=PROMPTS[:32],
        eval_prompts=EVAL_PROMPTS,
        config=config,
    )


if __name__ == "__main__":
    hparams = {} if len(sys.argv) == 1 else json.loads(sys.argv[1])
    main(hparams)
