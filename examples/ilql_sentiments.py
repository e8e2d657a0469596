"""ILQL sentiments (parity: reference examples/ilql_sentiments.py) — offline
version trains on synthetic reward-labeled review stubs."""

import json
import sys

import os as _os
import sys as _sys
_sys.path.insert(0, _os.path.dirname(_os.path.dirname(_os.path.abspath(__file__))))
import trlx_amd as trlx
from trlx_amd.data.default_configs import default_ilql_config
from trlx_amd.models.nn.config import preset

from ppo_sentiments import EVAL_PROMPTS, sentiment_reward

SAMPLES = [
    "The movie was great and I loved it",
    "The movie was terrible and boring",
    "I watched this film and it was amazing",
    "I watched this film and hated every minute",
    "This picture is a masterpiece",
    "This picture is a complete mess",
    "Overall the acting was superb",
    "Overall the acting felt poor and dull",
] * 16


def main(hparams={}):
    config = default_ilql_config()
    config.model.model_path = "gpt2"
    config.model.model_extra_configs = {"config": preset("gpt2").to_dict()}
    config.tokenizer.tokenizer_path = "byte"
    config.train.seq_length = 64
    config.train.batch_size = 32
    config = trlx.TRLConfig.update(config.to_dict(), hparams)

    rewards = sentiment_reward(SAMPLES)
    trlx.train(
        samples=SAMPLES,
        rewards=rewards,
        eval_prompts=EVAL_PROMPTS,
        metric_fn=lambda samples, **kw: {"sentiment": sentiment_reward(samples)},
        config=config,
    )


if __name__ == "__main__":
    hparams = {} if len(sys.argv) == 1 else json.loads(sys.argv[1])
    main(hparams)
