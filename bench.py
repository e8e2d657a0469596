#!/usr/bin/env python3
"""Flagship benchmark: PPO samples/sec, GPT-2-small "sentiments" shape.

``--method ilql`` instead measures offline ILQL training samples/sec
(BASELINE.json config #2: GPT-2-small ILQL sentiments, bf16, 1 GPU): one
step = one optimizer step over a batch of reward-labeled sequences +
periodic target-Q Polyak sync.

BASELINE.json metric: "PPO samples/sec (whole node), GPT-2 sentiments" with
the reference's canonical hyperparameters as the algorithmic anchor
(default_configs: num_rollouts=128, chunk_size=128, ppo_epochs=4, batch 32,
seq 1024, max_new_tokens=40).  One bench *step* = one full PPO outer cycle:
collect num_rollouts experiences (generate + reward + logprob/value/ref pass)
and run ppo_epochs optimization epochs over them — so samples/sec is the true
end-to-end RLHF pipeline throughput, nothing skipped.

Data: synthetic prompts over a model-vocab-sized synthetic vocabulary with perfect
decode/encode round-trip; weights: random-init GPT-2-small; reward: cheap
deterministic stateless function of the sample string, scored on EVERY rank
(method.local_rewards — the NeMo-path protocol; the gather-to-rank-0 +
scatter protocol is implemented and gloo-tested but serializes at DP=8).

Launch (the driver's contract):
  python bench.py --gpus N --steps K --warmup W
  torchrun --nnodes=1 --nproc-per-node N bench.py --gpus N --steps K --warmup W
"""

import argparse
import json
import os
import time

import torch


def model_config(args):
    from trlx_amd.models.nn.config import TransformerConfig, preset

    if getattr(args, "tiny_smoke", False):
        # CPU dry-run shape for the torchrun contract test — NOT a benchmark
        return TransformerConfig(vocab_size=512, hidden_size=64, num_layers=2, num_heads=2,
                                 max_position_embeddings=256, arch_name="gpt2")
    return preset(args.model)


def build_trainer(args, reward_fn=None):
    from trlx_amd.data.default_configs import default_ppo_config
    from trlx_amd.pipeline.offline_pipeline import PromptPipeline
    from trlx_amd.utils.loading import get_trainer

    config = default_ppo_config()
    config.model.model_path = args.model
    model_cfg = model_config(args)
    config.model.model_extra_configs = {"config": model_cfg.to_dict()}
    config.model.num_layers_unfrozen = args.num_layers_unfrozen
    # tokenizer vocab MUST match the model vocab: sampled/synthetic ids at or
    # above vocab_size index out of the embedding (llama V=32000 < the
    # default 50257 -> device memory fault)
    config.tokenizer.tokenizer_path = f"synthetic:{model_cfg.vocab_size}"
    config.train.seq_length = args.seq_len
    config.train.batch_size = args.batch_size
    config.train.total_steps = 10**9
    config.train.eval_interval = 10**9
    config.train.checkpoint_interval = 10**9
    config.train.tracker = None
    config.train.save_best = False
    config.method.num_rollouts = args.num_rollouts
    config.method.chunk_size = args.chunk_size
    config.method.ppo_epochs = args.ppo_epochs
    config.method.gen_kwargs = dict(
        max_new_tokens=args.max_new_tokens, top_k=0, top_p=1.0, do_sample=True
    )
    # the synthetic reward is stateless + deterministic: score on every rank
    # (the NeMo-path protocol) instead of serializing through rank 0 — at
    # DP=8 the gather->rank-0-score->scatter round trip idles 7 ranks
    config.method.local_rewards = True

    if reward_fn is None:
        def reward_fn(samples, prompts, outputs, **kwargs):
            # cheap deterministic stand-in for the sentiment classifier
            return [float((len(s) * 2654435761) % 1000) / 1000.0 - 0.5 for s in samples]

    trainer = get_trainer(config.train.trainer)(config=config, reward_fn=reward_fn)

    torch.manual_seed(1234 + int(os.environ.get("RANK", 0)))
    prompt_tokens = torch.randint(3, model_cfg.vocab_size,
                                  (args.num_prompts, args.prompt_len)).tolist()
    prompts = [" ".join(f"t{t}" for t in row) for row in prompt_tokens]
    pipeline = PromptPipeline(prompts, args.prompt_len + 2, trainer.tokenizer)
    trainer.add_prompt_pipeline(pipeline)
    return trainer, config


def run_cycle(trainer, config, phase_times=None):
    """One PPO outer cycle: experience collection + ppo_epochs optimization."""
    from trlx_amd.pipeline import MiniBatchIterator

    def tick():
        if phase_times is not None and torch.cuda.is_available():
            torch.cuda.synchronize()
        return time.time()

    t0 = tick()
    trainer.store.clear_history()
    trainer.make_experience(config.method.num_rollouts)
    t1 = tick()
    trainer.model.eval()  # learn()'s convention: PPO train-step forward sans dropout
    for _ in range(config.method.ppo_epochs):
        loader = trainer.store.create_loader(config.train.batch_size, shuffle=True)
        for minibatch in MiniBatchIterator(loader, trainer.mb_size, trainer.num_mb):
            for microbatch in minibatch:
                with trainer._accumulate():
                    # same body as NativeRLTrainer.learn: the captured
                    # fwd+bwd hipGraph when eligible, eager otherwise
                    if trainer._graphed_loss_backward(microbatch) is None:
                        loss, _ = trainer.loss(microbatch)
                        loss.backward()
            trainer.reducer.finalize()
            trainer.opt.step()
            trainer.opt.zero_grad()
            trainer.scheduler.step()
    t2 = tick()
    if phase_times is not None:
        phase_times.setdefault("experience", 0.0)
        phase_times.setdefault("train", 0.0)
        phase_times["experience"] += t1 - t0
        phase_times["train"] += t2 - t1


def parse_args(argv=None):
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=5)
    p.add_argument("--warmup", type=int, default=2)
    p.add_argument("--model", type=str, default="gpt2")
    p.add_argument("--seq-len", type=int, default=1024)
    p.add_argument("--prompt-len", type=int, default=64)
    p.add_argument("--max-new-tokens", type=int, default=40)
    p.add_argument("--batch-size", type=int, default=32)
    p.add_argument("--num-rollouts", type=int, default=128)
    p.add_argument("--chunk-size", type=int, default=128)
    p.add_argument("--ppo-epochs", type=int, default=4)
    p.add_argument("--num-layers-unfrozen", type=int, default=2)
    p.add_argument("--num-prompts", type=int, default=512)
    p.add_argument("--phases", action="store_true", help="print per-phase times (experience vs train)")
    p.add_argument("--method", choices=["ppo", "ilql"], default="ppo")
    p.add_argument("--tiny-smoke", action="store_true",
                   help="tiny model for the torchrun CPU contract dry-run (not a benchmark)")
    p.add_argument("--no-secondary", action="store_true",
                   help="skip the secondary ILQL / model-ladder measurements")
    return p.parse_args(argv)


def main():
    args = parse_args()

    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    rank = int(os.environ.get("RANK", 0))
    world = int(os.environ.get("WORLD_SIZE", 1))

    import trlx_amd  # noqa: F401  (also loads the HIP extension path)
    from trlx_amd.parallel import comm

    if world > 1:
        # first-contact sanity: verify the collective stack before training
        import sys

        comm.init_distributed()
        comm.preflight(log=lambda m: print(m, file=sys.stderr))

    if args.method == "ilql":
        return run_ilql(args)

    trainer, config = build_trainer(args)

    def sync():
        comm.barrier()
        if torch.cuda.is_available():
            torch.cuda.synchronize()

    for _ in range(args.warmup):
        run_cycle(trainer, config)

    sync()
    phase_times = {}
    t0 = time.time()
    for _ in range(args.steps):
        run_cycle(trainer, config, phase_times)
    sync()
    elapsed = time.time() - t0
    if args.phases and rank == 0:
        import sys
        per = {k: round(1000 * v / args.steps, 1) for k, v in phase_times.items()}
        print(f"phase ms/step: {per}", file=sys.stderr)
        exp = {k: round(1000 * float(v), 1) for k, v in
               getattr(trainer, "last_experience_stats", {}).items() if k.startswith("time/")}
        print(f"experience sub-phases ms: {exp}", file=sys.stderr)

    # max over ranks -> whole-job time
    if world > 1:
        t = torch.tensor([elapsed], dtype=torch.float64, device=comm.get_device())
        torch.distributed.all_reduce(t, torch.distributed.ReduceOp.MAX)
        elapsed = float(t.item())

    total_samples = args.num_rollouts * args.steps * world
    samples_per_sec = total_samples / elapsed

    if rank == 0:
        result = {
            "metric": "ppo_samples_per_sec",
            "value": round(samples_per_sec, 3),
            "unit": "samples/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(1000.0 * elapsed / args.steps, 2),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "bf16" if torch.cuda.is_available() else "fp32",
            "data": "synthetic prompts (model-sized synthetic vocab), random-init weights, deterministic synthetic reward",
            "config": {
                "model": args.model,
                "global_batch": args.batch_size * world,
                "seq_len": args.seq_len,
                "prompt_len": args.prompt_len,
                "max_new_tokens": args.max_new_tokens,
                "num_rollouts_per_gpu": args.num_rollouts,
                "ppo_epochs": args.ppo_epochs,
                "num_layers_unfrozen": args.num_layers_unfrozen,
                "parallelism": f"dp{world}",
                "local_rewards": True,
            },
            "phases_ms_per_step": {k: round(1000 * v / args.steps, 1)
                                   for k, v in phase_times.items()},
        }
        if torch.cuda.is_available():
            result["peak_mem_gb"] = round(torch.cuda.max_memory_allocated() / 2**30, 2)
    else:
        result = None

    # secondary measurements (1-GPU only): BASELINE config #2 (ILQL bf16) and
    # a model-ladder entry, carried inside the single JSON line so the driver
    # records more than the flagship number
    if world == 1 and not args.no_secondary and not args.tiny_smoke:
        del trainer
        if torch.cuda.is_available():
            torch.cuda.empty_cache()
        secondary = {}
        try:
            ilql = run_ilql(_clone_args(args, steps=min(args.steps, 5), warmup=1), emit=False)
            secondary["ilql"] = {k: ilql[k] for k in
                                 ("metric", "value", "unit", "ms_per_step", "steps", "config")}
        except Exception as e:  # secondary must never sink the flagship number
            secondary["ilql"] = {"error": str(e)}
        try:
            secondary["ladder_gpt2_xl"] = run_ladder(args)
        except Exception as e:
            secondary["ladder_gpt2_xl"] = {"error": str(e)}
        if result is not None:
            result["secondary"] = secondary

    if result is not None:
        print(json.dumps(result))


def _clone_args(args, **over):
    import copy

    a = copy.copy(args)
    for k, v in over.items():
        setattr(a, k, v)
    return a


def run_ladder(args):
    """A model-ladder entry: gpt2-xl (1.5B) random-init PPO, 2 measured
    cycles at the flagship hyperparameters."""
    a = _clone_args(args, model="gpt2-xl", steps=2, warmup=1)
    trainer, config = build_trainer(a)
    for _ in range(a.warmup):
        run_cycle(trainer, config)
    if torch.cuda.is_available():
        torch.cuda.synchronize()
    t0 = time.time()
    phase_times = {}
    for _ in range(a.steps):
        run_cycle(trainer, config, phase_times)
    if torch.cuda.is_available():
        torch.cuda.synchronize()
    elapsed = time.time() - t0
    del trainer
    if torch.cuda.is_available():
        torch.cuda.empty_cache()
    return {
        "metric": "ppo_samples_per_sec",
        "model": "gpt2-xl",
        "value": round(a.num_rollouts * a.steps / elapsed, 3),
        "unit": "samples/s",
        "steps": a.steps,
        "ms_per_step": round(1000.0 * elapsed / a.steps, 2),
        "phases_ms_per_step": {k: round(1000 * v / a.steps, 1) for k, v in phase_times.items()},
    }


def run_ilql(args, emit=True):
    import trlx_amd
    from trlx_amd.data.default_configs import default_ilql_config
    from trlx_amd.parallel import comm
    from trlx_amd.utils.loading import get_trainer

    rank = int(os.environ.get("RANK", 0))
    world = int(os.environ.get("WORLD_SIZE", 1))
    mcfg = model_config(args)
    config = default_ilql_config()
    config.model.model_path = args.model
    config.model.model_extra_configs = {"config": mcfg.to_dict()}
    config.tokenizer.tokenizer_path = f"synthetic:{mcfg.vocab_size}"
    config.train.seq_length = 64
    config.train.batch_size = 128
    config.train.tracker = None
    config.train.total_steps = 10**9
    trainer = get_trainer("ILQLTrainer")(config=config)

    torch.manual_seed(1234 + rank)
    # synthetic reward-labeled samples at the canonical ILQL shape (seq 64)
    toks = torch.randint(3, mcfg.vocab_size, (512, 60)).tolist()
    samples = [" ".join(f"t{t}" for t in row) for row in toks]
    rewards = [((i * 2654435761) % 1000) / 1000.0 - 0.5 for i in range(len(samples))]
    trainer.make_experience(samples, rewards, config.train.seq_length)
    loader = trainer.store.create_loader(config.train.batch_size)

    def sync():
        comm.barrier()
        if torch.cuda.is_available():
            torch.cuda.synchronize()

    def run_steps(n):
        it = iter(loader)
        for i in range(n):
            try:
                batch = next(it)
            except StopIteration:
                it = iter(loader)
                batch = next(it)
            loss, _ = trainer.loss(batch)
            trainer.model.train()
            loss.backward()
            trainer.reducer.finalize()
            trainer.opt.step()
            trainer.opt.zero_grad()
            if (i + 1) % config.method.steps_for_target_q_sync == 0:
                trainer.unwrapped_model.sync_target_q_heads()

    run_steps(args.warmup * 4)
    sync()
    t0 = time.time()
    run_steps(args.steps * 4)
    sync()
    elapsed = time.time() - t0
    if world > 1:
        t = torch.tensor([elapsed], dtype=torch.float64, device=comm.get_device())
        torch.distributed.all_reduce(t, torch.distributed.ReduceOp.MAX)
        elapsed = float(t.item())
    total = config.train.batch_size * args.steps * 4 * world
    result = None
    if rank == 0:
        result = {
            "metric": "ilql_samples_per_sec",
            "value": round(total / elapsed, 3),
            "unit": "samples/s",
            "n_gpus": world,
            "steps": args.steps * 4,
            "warmup": args.warmup * 4,
            "ms_per_step": round(1000.0 * elapsed / (args.steps * 4), 2),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "bf16" if torch.cuda.is_available() else "fp32",
            "data": "synthetic reward-labeled sequences (vocab 50257), random-init weights",
            "config": {"model": args.model, "global_batch": config.train.batch_size * world,
                       "seq_len": 64, "two_qs": True, "parallelism": f"dp{world}"},
        }
        if emit:
            print(json.dumps(result))
    return result


if __name__ == "__main__":
    main()
