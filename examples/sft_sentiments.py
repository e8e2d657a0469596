"""SFT sentiments (parity: reference examples/sft_sentiments.py) — fine-tune
on positive review stubs only."""

import json
import sys

import os as _os
import sys as _sys
_sys.path.insert(0, _os.path.dirname(_os.path.dirname(_os.path.abspath(__file__))))
import trlx_amd as trlx
from trlx_amd.data.default_configs import default_sft_config
from trlx_amd.models.nn.config import preset

from ppo_sentiments import EVAL_PROMPTS, sentiment_reward

POSITIVE_SAMPLES = [
    "The movie was great and I loved it",
    "I watched this film and it was amazing",
    "This picture is a masterpiece",
    "Overall the acting was superb and brilliant",
    "From the first scene it was wonderful",
] * 32


def main(hparams={}):
    config = default_sft_config()
    config.model.model_path = "gpt2"
    config.model.model_extra_configs = {"config": preset("gpt2").to_dict()}
    config.tokenizer.tokenizer_path = "byte"
    config.train.seq_length = 64
    config.train.batch_size = 32
    config = trlx.TRLConfig.update(config.to_dict(), hparams)

    trlx.train(
        samples=POSITIVE_SAMPLES,
        eval_prompts=EVAL_PROMPTS,
        metric_fn=lambda samples, **kw: {"sentiment": sentiment_reward(samples)},
        config=config,
    )


if __name__ == "__main__":
    hparams = {} if len(sys.argv) == 1 else json.loads(sys.argv[1])
    main(hparams)
