"""PPO sentiments with LoRA adapters (parity: reference
examples/ppo_sentiments_peft.py — PEFT fine-tuning where the frozen base
model doubles as the KL reference via adapter toggling)."""

import json
import sys

import os as _os
import sys as _sys
_sys.path.insert(0, _os.path.dirname(_os.path.dirname(_os.path.abspath(__file__))))
import trlx_amd as trlx
from trlx_amd.data.default_configs import default_ppo_config
from trlx_amd.models.nn.config import preset

from ppo_sentiments import EVAL_PROMPTS, PROMPTS, sentiment_reward


def default_config():
    config = default_ppo_config()
    config.model.model_path = "gpt2"
    config.model.model_extra_configs = {"config": preset("gpt2").to_dict()}
    config.model.peft_config = {
        "peft_type": "LORA", "r": 8, "lora_alpha": 32, "lora_dropout": 0.0,
    }
    config.tokenizer.tokenizer_path = "byte"
    config.train.seq_length = 128
    config.train.batch_size = 16
    config.method.chunk_size = 32
    config.method.num_rollouts = 64
    config.method.gen_kwargs = dict(max_new_tokens=24, top_k=0, top_p=1.0, do_sample=True)
    return config


def main(hparams={}):
    config = trlx.TRLConfig.update(default_config().to_dict(), hparams)
    trlx.train(reward_fn=sentiment_reward, prompts=PROMPTS, eval_prompts=EVAL_PROMPTS,
               config=config)


if __name__ == "__main__":
    main({} if len(sys.argv) == 1 else json.loads(sys.argv[1]))
