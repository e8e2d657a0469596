#!/usr/bin/env python3
"""Train-phase-only microbench: PPO optimizer steps on a fabricated store
(no generation), for isolating/profiling the training step."""
import argparse
import sys
import time

import torch

sys.path.insert(0, ".")
from bench import build_trainer  # noqa: E402


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--steps", type=int, default=32)
    p.add_argument("--warmup", type=int, default=8)
    p.add_argument("--batch-size", type=int, default=32)
    p.add_argument("--prompt-len", type=int, default=64)
    p.add_argument("--resp-len", type=int, default=41)
    p.add_argument("--model", type=str, default="gpt2")
    p.add_argument("--num-layers-unfrozen", type=int, default=2)
    args = p.parse_args()
    args.seq_len = 1024
    args.max_new_tokens = args.resp_len
    args.num_rollouts = args.batch_size
    args.chunk_size = args.batch_size
    args.ppo_epochs = 4
    args.num_prompts = 64

    from trlx_amd.data.ppo_types import PPORLElement

    trainer, config = build_trainer(args)
    torch.manual_seed(0)
    elems = [
        PPORLElement(
            query_tensor=torch.randint(3, 50257, (args.prompt_len,)),
            response_tensor=torch.randint(3, 50257, (args.resp_len,)),
            logprobs=torch.randn(args.resp_len) - 5,
            values=torch.randn(args.resp_len) * 0.1,
            rewards=torch.randn(args.resp_len) * 0.01,
        )
        for _ in range(args.batch_size)
    ]
    trainer.store.clear_history()
    trainer.store.push(elems)
    loader = trainer.store.create_loader(args.batch_size, shuffle=False)
    batch = next(iter(loader))

    def step():
        loss, _ = trainer.loss(batch)
        trainer.model.train()
        loss.backward()
        trainer.opt.step()
        trainer.opt.zero_grad()

    for _ in range(args.warmup):
        step()
    torch.cuda.synchronize() if torch.cuda.is_available() else None
    t0 = time.time()
    for _ in range(args.steps):
        step()
    torch.cuda.synchronize() if torch.cuda.is_available() else None
    dt = (time.time() - t0) / args.steps
    print(f"train step: {dt*1000:.2f} ms  ({args.batch_size * (args.prompt_len+args.resp_len)} tokens)")


if __name__ == "__main__":
    main()
