"""Stage 1/3 — supervised fine-tuning on (post, TL;DR summary) pairs.

Parity: reference examples/summarize_rlhf/sft/train_gptj_summarize.py (HF
Trainer on GPT-J + openai_summarize_tldr).  Offline analog: native SFT
trainer on the synthetic TL;DR task, byte tokenizer, saving an HF-format
checkpoint for the later stages.
"""

import json
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

import trlx_amd as trlx
from trlx_amd.data.default_configs import default_sft_config
from trlx_amd.models.nn.config import preset

from synthetic_tldr import make_prompts, make_sft_samples, oracle_reward

OUT_DIR = os.environ.get("TRLX_AMD_SUMMARIZE_DIR", "ckpts/summarize_rlhf")


def main(hparams={}):
    config = default_sft_config()
    config.model.model_path = "gpt2"
    config.model.model_extra_configs = {"config": preset("gpt2").to_dict()}
    config.tokenizer.tokenizer_path = "byte"
    config.train.seq_length = 64
    config.train.batch_size = 32
    config.train.epochs = 2
    config.train.checkpoint_dir = os.path.join(OUT_DIR, "sft")
    config = trlx.TRLConfig.update(config.to_dict(), hparams)

    trainer = trlx.train(
        samples=make_sft_samples(512),
        eval_prompts=make_prompts(16),
        metric_fn=lambda samples, **kw: {"oracle": oracle_reward(samples)},
        config=config,
    )
    hf_dir = os.path.join(config.train.checkpoint_dir, "hf_model")
    trainer.save_pretrained(hf_dir)
    return hf_dir


if __name__ == "__main__":
    hparams = {} if len(sys.argv) == 1 else json.loads(sys.argv[1])
    main(hparams)
