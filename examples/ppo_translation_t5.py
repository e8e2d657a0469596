"""PPO translation with T5 (parity: reference examples/ppo_translation_t5.py
— WMT de-en with BLEU reward).

Offline adaptation: a tiny built-in parallel "corpus" (string transforms)
and a character-overlap reward against the target — the same seq2seq PPO
machinery (T5 wrapper, right padding, decoder-side rollouts) as the real
task; point ``model_path`` at a local t5 dir + swap ``reward_fn`` for
sacrebleu to run WMT."""

import json
import sys

sys.path.insert(0, ".")
import trlx_amd as trlx
from trlx_amd.data.default_configs import default_ppo_config
from trlx_amd.models.nn.seq2seq import Seq2SeqConfig

T5_TINY = Seq2SeqConfig(vocab_size=500, d_model=64, d_kv=32, num_heads=2, d_ff=128,
                        num_layers=2, decoder_start_token_id=2, pad_token_id=2,
                        eos_token_id=1)

# toy deterministic "translation": reverse the word order
CORPUS = [("the cat sat", "sat cat the"), ("a dog runs fast", "fast runs dog a"),
          ("birds fly high", "high fly birds"), ("we like green tea", "tea green like we"),
          ("rain falls today", "today falls rain"), ("stars shine bright", "bright shine stars")]
TARGETS = {src: tgt for src, tgt in CORPUS}


def overlap_reward(samples, prompts, outputs, **kwargs):
    """Character-bigram F1 vs the target (stands in for BLEU offline)."""
    rs = []
    for prompt, out in zip(prompts, outputs):
        tgt = TARGETS.get(prompt.strip(), "")
        a = {out[i : i + 2] for i in range(len(out) - 1)}
        b = {tgt[i : i + 2] for i in range(len(tgt) - 1)}
        rs.append(2 * len(a & b) / max(len(a) + len(b), 1))
    return rs


def main(hparams={}):
    config = default_ppo_config()
    config.model.model_path = "t5-small"
    config.model.model_arch_type = "seq2seq"
    config.model.model_extra_configs = {"config": T5_TINY.to_dict()}
    config.model.num_layers_unfrozen = 2
    config.tokenizer.tokenizer_path = "byte"
    config.tokenizer.padding_side = "right"
    config.train.seq_length = 64
    config.train.batch_size = 8
    config.train.total_steps = 30
    config.method.chunk_size = 8
    config.method.num_rollouts = 16
    config.method.gen_kwargs = dict(max_new_tokens=16, top_k=0, top_p=1.0, do_sample=True)
    config = trlx.TRLConfig.update(config.to_dict(), hparams)

    prompts = [src for src, _ in CORPUS] * 4
    trlx.train(
        reward_fn=overlap_reward,
        prompts=prompts,
        eval_prompts=[src for src, _ in CORPUS[:2]],
        config=config,
    )


if __name__ == "__main__":
    hparams = {} if len(sys.argv) == 1 else json.loads(sys.argv[1])
    main(hparams)
