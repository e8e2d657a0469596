"""PPO on the grounded program-synthesis DSL (parity: reference
examples/experiments/grounded_program_synthesis/train_trlx.py): the model
completes "Input: ... Output: ... Function:" prompts with a program; reward
comes from RUNNING the program through the interpreter."""

import json
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

import trlx_amd as trlx  # noqa: E402
from lang import make_dataset, reward_fn  # noqa: E402
from trlx_amd.data.default_configs import default_ppo_config  # noqa: E402
from trlx_amd.models.nn.config import preset  # noqa: E402


def default_config():
    config = default_ppo_config()
    config.model.model_path = "gpt2"
    config.model.model_extra_configs = {"config": preset("gpt2").to_dict()}
    config.model.num_layers_unfrozen = 2
    config.tokenizer.tokenizer_path = "byte"
    config.train.seq_length = 160
    config.train.batch_size = 16
    config.method.chunk_size = 16
    config.method.num_rollouts = 32
    config.method.gen_kwargs = dict(max_new_tokens=32, top_k=0, top_p=1.0, do_sample=True)
    return config


def main(hparams={}):
    config = trlx.TRLConfig.update(default_config().to_dict(), hparams)
    prompts = [p for p, _ in make_dataset(256)]
    trlx.train(reward_fn=reward_fn, prompts=prompts, eval_prompts=prompts[:8], config=config)


if __name__ == "__main__":
    main({} if len(sys.argv) == 1 else json.loads(sys.argv[1]))
