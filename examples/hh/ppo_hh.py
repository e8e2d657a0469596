"""PPO on Helpful-Harmless dialogues (parity: reference examples/hh/ppo_hh.py).

Reference behavior reproduced offline:
- CONFIG_NAME env selects the model size tier (reference ppo_hh.py:70 —
  125M/1B/6B/20B; here the matching presets with random init, no network);
- reward scoring goes OUT-OF-BAND to a reward server over HTTP when
  TRLX_AMD_REWARD_URL is set (reference used Triton gRPC,
  ppo_hh.py:109-120); otherwise the in-process oracle scores directly;
- dialogue prompts in the "\\n\\nHuman: ... \\n\\nAssistant:" format.

    python reward_server.py --port 8710 &            # optional
    TRLX_AMD_REWARD_URL=http://127.0.0.1:8710 python ppo_hh.py
"""

import json
import os
import sys
from typing import List

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

import trlx_amd as trlx
from trlx_amd.data.default_configs import default_ppo_config
from trlx_amd.models.nn.config import preset

from hh_task import make_prompts, oracle_reward

CONFIG_SIZES = {"125M": "gpt2", "1B": "gpt2-xl", "6B": "gptj", "20B": "gpt-neox"}


def preset_for_config_name():
    name = os.environ.get("CONFIG_NAME", "125M")
    return CONFIG_SIZES.get(name, "gpt2")


def http_reward_fn(url: str):
    import requests

    def reward_fn(samples: List[str], **kwargs):
        r = requests.post(f"{url}/reward", json={"samples": samples}, timeout=60)
        r.raise_for_status()
        return r.json()["scores"]

    return reward_fn


def main(hparams={}):
    model = preset_for_config_name()
    config = default_ppo_config()
    config.model.model_path = model
    config.model.model_extra_configs = {"config": preset(model).to_dict()}
    config.model.num_layers_unfrozen = 2
    config.tokenizer.tokenizer_path = "byte"
    config.train.seq_length = 128
    config.train.batch_size = 8
    config.method.chunk_size = 16
    config.method.num_rollouts = 32
    config.method.gen_kwargs = dict(max_new_tokens=24, top_k=0, top_p=1.0, do_sample=True)
    config = trlx.TRLConfig.update(config.to_dict(), hparams)

    url = os.environ.get("TRLX_AMD_REWARD_URL")
    reward_fn = http_reward_fn(url) if url else (lambda samples, **kw: oracle_reward(samples))

    trlx.train(
        reward_fn=reward_fn,
        prompts=make_prompts(128),
        eval_prompts=make_prompts(16, seed=9),
        metric_fn=lambda samples, **kw: {"oracle": oracle_reward(samples)},
        config=config,
    )


if __name__ == "__main__":
    hparams = {} if len(sys.argv) == 1 else json.loads(sys.argv[1])
    main(hparams)
