"""PPO with dense per-token rewards (parity: reference
examples/ppo_dense_sentiments.py) — reward_fn returns one score per response
token instead of a scalar, exercising the dense-reward path of
make_experience."""

import json
import sys

import os as _os
import sys as _sys
_sys.path.insert(0, _os.path.dirname(_os.path.dirname(_os.path.abspath(__file__))))
import trlx_amd as trlx

from ppo_sentiments import EVAL_PROMPTS, PROMPTS, default_config, sentiment_reward


def dense_reward(samples, prompts, outputs, tokenizer, **kwargs):
    """Spread the sample-level sentiment score over the response tokens,
    weighted toward the end (the reference uses per-token classifier deltas)."""
    scores = sentiment_reward(samples)
    out = []
    for score, output in zip(scores, outputs):
        n = max(len(tokenizer(output).input_ids), 1)
        per_tok = [score / n] * n
        out.append(per_tok)
    return out


def main(hparams={}):
    config = trlx.TRLConfig.update(default_config().to_dict(), hparams)
    trlx.train(
        reward_fn=dense_reward,
        prompts=PROMPTS,
        eval_prompts=EVAL_PROMPTS,
        config=config,
    )


if __name__ == "__main__":
    hparams = {} if len(sys.argv) == 1 else json.loads(sys.argv[1])
    main(hparams)
