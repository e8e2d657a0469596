"""Stage 3/3 — PPO against the trained reward model.

Parity: reference examples/summarize_rlhf/trlx_gptj_text_summarization.py:
load the SFT policy and the stage-2 reward model, score rollouts with the RM
normalized against the gold summary's RM score (reference :147-153), train
with trlx.train(reward_fn=...).
"""

import json
import os
import sys
from typing import List

import torch

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

import trlx_amd as trlx
from trlx_amd.data.default_configs import default_ppo_config
from trlx_amd.models.nn.config import preset
from trlx_amd.utils.tokenizer import ByteTokenizer

from reward_model import RewardModel
from synthetic_tldr import SEP, make_post, make_prompts, oracle_reward

OUT_DIR = os.environ.get("TRLX_AMD_SUMMARIZE_DIR", "ckpts/summarize_rlhf")


def main(hparams={}, sft_dir=None, rm_dir=None):
    sft_dir = sft_dir or os.path.join(OUT_DIR, "sft", "hf_model")
    rm_dir = rm_dir or os.path.join(OUT_DIR, "rm")
    device = torch.device("cuda" if torch.cuda.is_available() else "cpu")
    tok = ByteTokenizer()

    config = default_ppo_config()
    config.model.model_path = sft_dir if os.path.isdir(sft_dir) else "gpt2"
    if not os.path.isdir(sft_dir):
        config.model.model_extra_configs = {"config": preset("gpt2").to_dict()}
    config.tokenizer.tokenizer_path = "byte"
    config.train.seq_length = 64
    config.train.batch_size = 8
    config.method.chunk_size = 16
    config.method.num_rollouts = 32
    config.method.init_kl_coef = 0.05
    config.method.gen_kwargs = dict(max_new_tokens=8, top_k=0, top_p=1.0, do_sample=True)
    config = trlx.TRLConfig.update(config.to_dict(), hparams)

    rw_model = RewardModel.load_checkpoint(rm_dir, device=device)
    rw_model.eval()
    if device.type == "cuda":
        rw_model.transformer.to(torch.bfloat16)
        rw_model.v_head.to(torch.float32)

    # gold reference continuations for score normalization (reference
    # trlx_gptj_text_summarization.py:171-177 post_summary_dict)
    post_summary = {}
    import random
    rng = random.Random(2)
    for _ in range(256):
        post, gold = make_post(rng)
        post_summary[post] = gold

    def rm_scores(samples: List[str]) -> torch.Tensor:
        ids = [tok._encode_one(s, config.train.seq_length, truncation=True) + [tok.eos_token_id]
               for s in samples]
        width = max(len(i) for i in ids)
        ids = [i + [tok.pad_token_id] * (width - len(i)) for i in ids]
        with torch.no_grad():
            return rw_model.score(torch.tensor(ids, dtype=torch.long, device=device)).cpu()

    def reward_fn(samples: List[str], **kwargs) -> torch.Tensor:
        originals = []
        for s in samples:
            post = s.split(SEP)[0] + SEP if SEP in s else s
            gold = post_summary.get(post)
            if gold is None:
                # prompt not in the table (truncated decode): gold = topic word
                words = post.strip().split()
                gold = words[0] if words else ""
            originals.append(post + " " + gold)
        return rm_scores(samples) - rm_scores(originals)

    prompts = [p for p in post_summary.keys()]
    trainer = trlx.train(
        reward_fn=reward_fn,
        prompts=prompts,
        eval_prompts=make_prompts(16, seed=9),
        metric_fn=lambda samples, **kw: {"oracle": oracle_reward(samples)},
        config=config,
    )
    return trainer


if __name__ == "__main__":
    hparams = {} if len(sys.argv) == 1 else json.loads(sys.argv[1])
    main(hparams)
