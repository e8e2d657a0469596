"""ILQL with a T5 seq2seq model (parity: reference
examples/ilql_sentiments_t5.py).

Offline adaptation: random-init tiny-T5 + lexicon sentiment reward over
(prompt, continuation) samples; point ``model_path`` at a local flan-t5
directory for the real task."""

import json
import sys

sys.path.insert(0, ".")
import trlx_amd as trlx
from trlx_amd.data.default_configs import default_ilql_config
from trlx_amd.models.nn.seq2seq import Seq2SeqConfig

from ilql_sentiments import SAMPLES, sentiment_reward
from ppo_sentiments import EVAL_PROMPTS

T5_TINY = Seq2SeqConfig(vocab_size=500, d_model=64, d_kv=32, num_heads=2, d_ff=128,
                        num_layers=2, decoder_start_token_id=2, pad_token_id=2,
                        eos_token_id=1)


def main(hparams={}):
    config = default_ilql_config()
    config.model.model_path = "t5-small"
    config.model.model_arch_type = "seq2seq"
    config.model.model_extra_configs = {"config": T5_TINY.to_dict()}
    config.tokenizer.tokenizer_path = "byte"
    config.tokenizer.padding_side = "right"
    config.train.seq_length = 64
    config.train.batch_size = 16
    config.train.total_steps = 40
    config = trlx.TRLConfig.update(config.to_dict(), hparams)

    # seq2seq ILQL consumes (prompt, output) pairs; split the causal samples
    pairs = [(s[: len(s) // 2], s[len(s) // 2 :]) for s in SAMPLES]
    rewards = sentiment_reward(SAMPLES)
    trlx.train(
        samples=pairs,
        rewards=rewards,
        eval_prompts=EVAL_PROMPTS,
        config=config,
    )


if __name__ == "__main__":
    hparams = {} if len(sys.argv) == 1 else json.loads(sys.argv[1])
    main(hparams)
