"""Generate from a trained checkpoint (parity: reference
examples/nemo_ilql_inference.py / nemo_ppo_inference.py — load megatron
checkpoints and sample; here the native equivalents).

Loads either a `save_pretrained` HF-format directory (single-process) or a
trainer checkpoint dir containing `mp_rank_XX` TP shards (merged on load),
then streams completions for prompts from argv/stdin through the
hipGraph decode engine.

    python examples/inference_checkpoint.py --model ckpts/best_checkpoint/hf_model \
        --max-new-tokens 48 "prompt one" "prompt two"
"""

import sys
from argparse import ArgumentParser

sys.path.insert(0, ".")
import torch

from trlx_amd.models.modeling_base import PreTrainedModelWrapper
from trlx_amd.models.nn.generation import GenerateConfig, generate
from trlx_amd.utils.tokenizer import get_tokenizer


def main(argv=None):
    p = ArgumentParser()
    p.add_argument("--model", required=True,
                   help="save_pretrained dir (hf_model) or mp_rank_XX checkpoint dir")
    p.add_argument("--tokenizer", default=None)
    p.add_argument("--max-new-tokens", type=int, default=48)
    p.add_argument("--temperature", type=float, default=1.0)
    p.add_argument("--greedy", action="store_true")
    p.add_argument("prompts", nargs="*")
    args = p.parse_args(argv)

    model = PreTrainedModelWrapper.from_pretrained(args.model)
    base = model.base_model if hasattr(model, "base_model") else model
    tok = get_tokenizer(args.tokenizer or args.model)
    device = "cuda" if torch.cuda.is_available() else "cpu"
    base = base.to(device).eval()
    if device == "cuda":
        base.to(torch.bfloat16)

    prompts = args.prompts or [line.strip() for line in sys.stdin if line.strip()]
    tok.padding_side = "left"
    enc = tok(prompts, padding=True, return_tensors="pt")
    ids = enc["input_ids"].to(device)
    mask = enc["attention_mask"].to(device)
    gen = GenerateConfig(max_new_tokens=args.max_new_tokens,
                         do_sample=not args.greedy, temperature=args.temperature,
                         eos_token_id=tok.eos_token_id, pad_token_id=tok.pad_token_id)
    with torch.no_grad():
        out = generate(base, ids, mask, gen=gen)
    for prompt, row in zip(prompts, out):
        completion = tok.decode(row[ids.shape[1]:], skip_special_tokens=True)
        print(f"=== {prompt!r}\n{completion}\n")


if __name__ == "__main__":
    main()
