"""Architext (parity: reference examples/architext.py — optimize textual
interior-design layouts toward the fewest rooms).

Offline adaptation: the reference fine-tunes architext/gptj-162M; with no
network this runs a random-init GPT-2 preset on the same prompts with the
same programmatic reward (negative room count — each room in a layout string
is introduced by a ``:``).
"""

import json
import sys

import os as _os
import sys as _sys
_sys.path.insert(0, _os.path.dirname(_os.path.dirname(_os.path.abspath(__file__))))
import trlx_amd as trlx
from trlx_amd.data.default_configs import default_ppo_config
from trlx_amd.models.nn.config import preset


def reward_fn(samples, **kwargs):
    """Negative count of rooms for each sample (reference architext.py:7-9)."""
    return [-float(sample.count(":")) for sample in samples]


PROMPTS = [
    "[prompt] the bedroom is adjacent to the living room [layout]",
    "[prompt] a bedroom is adjacent to the living room [layout]",
    "[prompt] the bedroom is adjacent to the kitchen [layout]",
    "[prompt] a bedroom is adjacent to the kitchen [layout]",
    "[prompt] the kitchen is adjacent to the bathroom [layout]",
    "[prompt] a bathroom is adjacent to the living room [layout]",
    "[prompt] the bathroom is adjacent to the living room [layout]",
    "[prompt] the bedroom is not adjacent to the living room [layout]",
    "[prompt] a bedroom is not adjacent to the kitchen [layout]",
    "[prompt] the kitchen is not adjacent to the bathroom [layout]",
]


def default_config():
    config = default_ppo_config()
    config.model.model_path = "gpt2"
    config.model.model_extra_configs = {"config": preset("gpt2").to_dict()}
    config.model.num_layers_unfrozen = 2
    config.tokenizer.tokenizer_path = "byte"
    config.train.seq_length = 128
    config.train.batch_size = 16
    config.method.chunk_size = 16
    config.method.num_rollouts = 32
    config.method.gen_kwargs = dict(max_new_tokens=24, top_k=0, top_p=1.0, do_sample=True)
    return config


def main(hparams={}):
    config = trlx.TRLConfig.update(default_config().to_dict(), hparams)
    trlx.train(reward_fn=reward_fn, prompts=PROMPTS, eval_prompts=PROMPTS[:4], config=config)


if __name__ == "__main__":
    main({} if len(sys.argv) == 1 else json.loads(sys.argv[1]))
