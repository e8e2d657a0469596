"""PPO sentiments (parity: reference examples/ppo_sentiments.py — GPT-2 +
sentiment reward on IMDB prompts).

Offline adaptation: with no network, prompts are synthetic movie-review
stubs and the reward is a lexicon-based sentiment score instead of the
distilbert classifier; point ``model_path``/``tokenizer_path`` at local HF
directories to run the real thing.
"""

import json
import sys

import os as _os
import sys as _sys
_sys.path.insert(0, _os.path.dirname(_os.path.dirname(_os.path.abspath(__file__))))
import trlx_amd as trlx
from trlx_amd.data.default_configs import default_ppo_config
from trlx_amd.models.nn.config import preset

POSITIVE = {"good", "great", "best", "love", "loved", "amazing", "fun", "enjoy", "beautiful",
            "brilliant", "superb", "wonderful", "masterpiece", "excellent"}
NEGATIVE = {"bad", "worst", "hate", "hated", "boring", "awful", "terrible", "poor",
            "disappointing", "mess", "dull", "horrible", "waste"}


def sentiment_reward(samples, **kwargs):
    """Lexicon stand-in for the distilbert-imdb classifier (offline)."""
    scores = []
    for s in samples:
        words = s.lower().split()
        pos = sum(w.strip(".,!?") in POSITIVE for w in words)
        neg = sum(w.strip(".,!?") in NEGATIVE for w in words)
        scores.append(float(pos - neg))
    return scores


PROMPTS = [
    "The movie was", "I watched this film and", "This picture is", "Overall the acting",
    "The director clearly", "From the first scene", "My favorite part", "The plot",
] * 16

EVAL_PROMPTS = ["The movie was", "I watched this film and"] * 16


def main(hparams={}):
    config = trlx.TRLConfig.update(default_config().to_dict(), hparams)
    trlx.train(
        reward_fn=sentiment_reward,
        prompts=PROMPTS,
        eval_prompts=EVAL_PROMPTS,
        config=config,
    )


def default_config():
    config = default_ppo_config()
    config.model.model_path = "gpt2"
    config.model.model_extra_configs = {"config": preset("gpt2").to_dict()}
    config.model.num_layers_unfrozen = 2
    config.tokenizer.tokenizer_path = "byte"
    config.train.seq_length = 128
    config.train.batch_size = 16
    config.method.chunk_size = 32
    config.method.num_rollouts = 64
    config.method.gen_kwargs = dict(max_new_tokens=24, top_k=0, top_p=1.0, do_sample=True)
    return config


if __name__ == "__main__":
    hparams = {} if len(sys.argv) == 1 else json.loads(sys.argv[1])
    main(hparams)
