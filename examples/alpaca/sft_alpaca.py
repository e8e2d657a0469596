"""SFT on Alpaca-format instruction data (parity: reference
examples/alpaca/sft_alpaca.py).

Offline adaptation: ``--data`` takes a local JSON list of
{"instruction", "input", "output"} records (the alpaca_data.json schema);
without it a tiny built-in set keeps the script runnable end-to-end.
Dialog pairs go through DialogStore so the loss masks the prompt region,
like the reference's prompt/output split.
"""

import json
import os
import sys
from argparse import ArgumentParser

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.dirname(os.path.abspath(__file__)))))
import trlx_amd as trlx
from trlx_amd.data.default_configs import default_sft_config
from trlx_amd.models.nn.config import TransformerConfig

BUILTIN = [
    {"instruction": "Give three tips for staying healthy.", "input": "",
     "output": "Eat a balanced diet. Exercise regularly. Sleep well."},
    {"instruction": "Translate to French", "input": "good morning",
     "output": "bonjour"},
    {"instruction": "What is the capital of France?", "input": "",
     "output": "Paris."},
    {"instruction": "Summarize", "input": "The quick brown fox jumps over the lazy dog.",
     "output": "A fox jumps over a dog."},
] * 8


def preprocess(rec):
    """Alpaca prompt format (matches reference preprocess())."""
    if rec.get("input"):
        prompt = ("Below is an instruction that describes a task, paired with an input "
                  "that provides further context. Write a response that appropriately "
                  "completes the request.\n\n"
                  f"### Instruction:\n{rec['instruction']}\n\n"
                  f"### Input:\n{rec['input']}\n\n### Response:\n")
    else:
        prompt = ("Below is an instruction that describes a task. Write a response "
                  "that appropriately completes the request.\n\n"
                  f"### Instruction:\n{rec['instruction']}\n\n### Response:\n")
    return (prompt, rec["output"])


def main(argv=None):
    p = ArgumentParser()
    p.add_argument("--data", default=None, help="path to alpaca_data.json")
    p.add_argument("--hparams", default="{}")
    args = p.parse_args(argv)
    records = json.load(open(args.data)) if args.data else BUILTIN
    samples = [preprocess(r) for r in records]

    config = default_sft_config()
    config.model.model_path = "gpt2"
    tiny = TransformerConfig(vocab_size=500, hidden_size=64, num_layers=2, num_heads=2,
                             max_position_embeddings=512, arch_name="gpt2")
    config.model.model_extra_configs = {"config": tiny.to_dict()}
    config.tokenizer.tokenizer_path = "byte"
    config.train.seq_length = 256
    config.train.batch_size = 4
    config.train.total_steps = 20
    config.train.eval_interval = 20
    config.train.checkpoint_interval = 10**9
    config.train.tracker = None
    config = trlx.TRLConfig.update(config.to_dict(), json.loads(args.hparams))

    trlx.train(
        samples=samples,
        eval_prompts=[preprocess(r)[0] for r in records[:2]],
        config=config,
    )


if __name__ == "__main__":
    main(sys.argv[1:])
