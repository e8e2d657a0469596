"""Stage 2/3 — train the pairwise reward model on synthetic comparisons.

Parity: reference examples/summarize_rlhf/reward_model/train_reward_model_gptj.py
(HF Trainer + DeepSpeed on GPT-J comparisons).  Offline analog: a plain
training loop on the native transformer — FusedAdamW (arena HIP kernel) on
GPU, torch AdamW on CPU — reporting pairwise accuracy on held-out pairs.
"""

import argparse
import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

from trlx_amd.utils.tokenizer import ByteTokenizer

from reward_model import RewardModel
from synthetic_tldr import make_comparisons

OUT_DIR = os.environ.get("TRLX_AMD_SUMMARIZE_DIR", "ckpts/summarize_rlhf")


def encode_pairs(pairs, tok, max_length=64):
    """Right-padded [2B, T]: chosen rows first, then rejected (the layout the
    RM forward expects, reference reward_model.py:52-57)."""
    texts = [c for c, _ in pairs] + [r for _, r in pairs]
    ids = [tok._encode_one(t, max_length, truncation=True) + [tok.eos_token_id] for t in texts]
    width = max(len(i) for i in ids)
    ids = [i + [tok.pad_token_id] * (width - len(i)) for i in ids]
    return torch.tensor(ids, dtype=torch.long)


def evaluate(model, pairs, tok, device, batch_size=32):
    correct = total = 0
    for i in range(0, len(pairs), batch_size):
        ids = encode_pairs(pairs[i:i + batch_size], tok).to(device)
        with torch.no_grad():
            out = model(ids)
        correct += int((out["chosen_end_scores"] > out["rejected_end_scores"]).sum())
        total += ids.shape[0] // 2
    return correct / max(total, 1)


def main(sft_dir=None, n_pairs=1024, epochs=2, batch_size=32, lr=1e-4, out_dir=None,
         config=None, seq_length=64):
    device = torch.device("cuda" if torch.cuda.is_available() else "cpu")
    tok = ByteTokenizer()
    base = config if config is not None else (
        sft_dir if sft_dir and os.path.isdir(sft_dir) else "gpt2")
    model = RewardModel.from_pretrained(base, tok.pad_token_id).to(device)
    if device.type == "cuda":
        model.transformer.to(torch.bfloat16)
        model.v_head.to(torch.float32)

    train_pairs = make_comparisons(n_pairs, seed=1)
    val_pairs = make_comparisons(max(n_pairs // 8, 16), seed=7)

    if device.type == "cuda":
        from trlx_amd.parallel.optim import FusedAdamW

        opt = FusedAdamW(model.parameters(), lr=lr, weight_decay=0.01)
    else:
        opt = torch.optim.AdamW(model.parameters(), lr=lr, weight_decay=0.01)

    for epoch in range(epochs):
        for i in range(0, len(train_pairs), batch_size):
            ids = encode_pairs(train_pairs[i:i + batch_size], tok, seq_length).to(device)
            out = model(ids)
            opt.zero_grad()
            out["loss"].backward()
            opt.step()
        acc = evaluate(model, val_pairs, tok, device)
        print(f"[rm] epoch {epoch}: loss {out['loss'].item():.4f}  val pairwise acc {acc:.3f}")

    out_dir = out_dir or os.path.join(OUT_DIR, "rm")
    model.float().save_checkpoint(out_dir)
    print(f"[rm] saved to {out_dir}")
    return out_dir, acc


if __name__ == "__main__":
    p = argparse.ArgumentParser()
    p.add_argument("--sft-dir", default=os.path.join(OUT_DIR, "sft", "hf_model"))
    p.add_argument("--pairs", type=int, default=1024)
    p.add_argument("--epochs", type=int, default=2)
    args = p.parse_args()
    main(sft_dir=args.sft_dir, n_pairs=args.pairs, epochs=args.epochs)
