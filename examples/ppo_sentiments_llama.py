"""PPO sentiments, Llama variant (parity: reference
examples/ppo_sentiments_llama.py — the same task on a Llama architecture:
RMSNorm + RoPE + SwiGLU exercise the llama-family kernels).

Offline adaptation: random-init tiny-llama preset + lexicon reward; point
``model_path`` at a local Llama HF directory to run the real model.
"""

import json
import sys

import os as _os
import sys as _sys
_sys.path.insert(0, _os.path.dirname(_os.path.dirname(_os.path.abspath(__file__))))
import trlx_amd as trlx
from trlx_amd.data.default_configs import default_ppo_config
from trlx_amd.models.nn.config import TransformerConfig

from ppo_sentiments import EVAL_PROMPTS, PROMPTS, sentiment_reward


def default_config():
    config = default_ppo_config()
    config.model.model_path = "llama"
    tiny_llama = TransformerConfig(
        vocab_size=2048, hidden_size=256, num_layers=4, num_heads=4,
        intermediate_size=688, max_position_embeddings=512, arch_name="llama",
        norm="rmsnorm", position_encoding="rope", activation="silu", swiglu=True,
        attn_bias=False, mlp_bias=False, tie_word_embeddings=False,
    )
    config.model.model_extra_configs = {"config": tiny_llama.to_dict()}
    config.model.num_layers_unfrozen = 2
    config.tokenizer.tokenizer_path = "byte"
    config.train.seq_length = 128
    config.train.batch_size = 16
    config.optimizer.kwargs["lr"] = 1e-5
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
