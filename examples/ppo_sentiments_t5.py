"""PPO with a T5 seq2seq model (parity: reference examples/ppo_sentiments_t5.py).

Offline adaptation: random-init tiny-T5 + lexicon sentiment reward; point
``model_path`` at a local flan-t5 directory for the real task.
"""

import json
import sys

import os as _os
import sys as _sys
_sys.path.insert(0, _os.path.dirname(_os.path.dirname(_os.path.abspath(__file__))))
import trlx_amd as trlx
from trlx_amd.data.default_configs import default_ppo_config
from trlx_amd.models.nn.seq2seq import Seq2SeqConfig

from ppo_sentiments import EVAL_PROMPTS, PROMPTS, sentiment_reward

T5_SMALL_LIKE = Seq2SeqConfig(vocab_size=32128, d_model=512, d_kv=64, num_heads=8, d_ff=2048,
                              num_layers=6, decoder_start_token_id=2, pad_token_id=2,
                              eos_token_id=1)


def main(hparams={}):
    config = default_ppo_config()
    config.model.model_path = "t5-small"
    config.model.model_arch_type = "seq2seq"
    config.model.model_extra_configs = {"config": T5_SMALL_LIKE.to_dict()}
    config.model.num_layers_unfrozen = 2
    config.tokenizer.tokenizer_path = "byte"
    config.tokenizer.padding_side = "right"
    config.train.seq_length = 128
    config.train.batch_size = 8
    config.method.chunk_size = 16
    config.method.num_rollouts = 32
    config.method.gen_kwargs = dict(max_new_tokens=24, top_k=0, top_p=1.0, do_sample=True)
    config = trlx.TRLConfig.update(config.to_dict(), hparams)

    trlx.train(
        reward_fn=sentiment_reward,
        prompts=PROMPTS,
        eval_prompts=EVAL_PROMPTS,
        config=config,
    )


if __name__ == "__main__":
    hparams = {} if len(sys.argv) == 1 else json.loads(sys.argv[1])
    main(hparams)
