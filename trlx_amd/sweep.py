"""Hyperparameter sweep runner: ``python -m trlx_amd.sweep config.yml script.py``.

Parity target: reference trlx/sweep.py (Ray Tune + W&B).  Ray and W&B are not
available offline, so this implements the same YAML search-space grammar
(strategy: grid / random / bayes-fallback-to-random) with a local
subprocess-per-trial executor and a jsonl results summary.  Trials run
sequentially (or torchrun-launched for multi-GPU) and report the tracker's
last logged metrics.

Search-space grammar (same keys as the reference's YAML):
  tune_config:
    mode: max | min
    metric: reward/mean
    search_alg: grid | random
    num_samples: 16            # random only
  method.init_kl_coef:
    strategy: loguniform
    values: [1e-4, 1e-1]
  train.seq_length:
    strategy: choice
    values: [512, 1024]
"""

import argparse
import itertools
import json
import os
import random
import subprocess
import sys
import time
from typing import Any, Dict, List


def parse_space(config: Dict[str, Any]):
    tune_config = config.pop("tune_config", {})
    dims = {}
    for param, spec in config.items():
        strategy = spec["strategy"]
        values = spec["values"]
        dims[param] = (strategy, values)
    return tune_config, dims


def sample_param(strategy: str, values: List[Any], rng: random.Random):
    if strategy == "choice":
        return rng.choice(values)
    if strategy == "uniform":
        return rng.uniform(values[0], values[1])
    if strategy == "loguniform":
        import math

        return math.exp(rng.uniform(math.log(values[0]), math.log(values[1])))
    if strategy == "quniform":
        lo, hi, q = values
        return round(rng.uniform(lo, hi) / q) * q
    raise ValueError(f"Unknown strategy: {strategy}")


def generate_trials(tune_config: Dict, dims: Dict, seed: int = 0):
    alg = tune_config.get("search_alg", "grid")
    if alg == "grid":
        grid_dims = {}
        for param, (strategy, values) in dims.items():
            if strategy != "choice":
                raise ValueError("grid search requires 'choice' strategies")
            grid_dims[param] = values
        keys = list(grid_dims)
        for combo in itertools.product(*grid_dims.values()):
            yield dict(zip(keys, combo))
    else:  # random / bayesopt fallback
        rng = random.Random(seed)
        n = int(tune_config.get("num_samples", 8))
        for _ in range(n):
            yield {p: sample_param(s, v, rng) for p, (s, v) in dims.items()}


def read_last_metrics(logging_dir: str) -> Dict[str, float]:
    path = os.path.join(logging_dir, "metrics.jsonl")
    last = {}
    if os.path.exists(path):
        with open(path) as f:
            for line in f:
                try:
                    last.update(json.loads(line))
                except json.JSONDecodeError:
                    continue
    return last


def main():
    parser = argparse.ArgumentParser(description="trlx_amd hyperparameter sweep")
    parser.add_argument("config", help="sweep YAML (search space + tune_config)")
    parser.add_argument("script", help="training script taking JSON hparams as argv[1]")
    parser.add_argument("--num-gpus", type=int, default=1, help="GPUs per trial")
    parser.add_argument("--output", default="sweep_results", help="results directory")
    parser.add_argument("--seed", type=int, default=0)
    args = parser.parse_args()

    import yaml

    with open(args.config) as f:
        space = yaml.safe_load(f)
    tune_config, dims = parse_space(space)
    metric = tune_config.get("metric", "reward/mean")
    mode = tune_config.get("mode", "max")

    os.makedirs(args.output, exist_ok=True)
    results = []
    for i, hparams in enumerate(generate_trials(tune_config, dims, args.seed)):
        trial_dir = os.path.join(args.output, f"trial_{i:03d}")
        hparams = dict(hparams)
        hparams["train.logging_dir"] = trial_dir
        hparams["train.tracker"] = "jsonl"
        hparams["train.checkpoint_dir"] = os.path.join(trial_dir, "ckpts")
        print(f"[sweep] trial {i}: {hparams}")
        if args.num_gpus > 1:
            cmd = [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
                   f"--nproc-per-node={args.num_gpus}", "--master-addr", "127.0.0.1",
                   args.script, json.dumps(hparams)]
        else:
            cmd = [sys.executable, args.script, json.dumps(hparams)]
        t0 = time.time()
        proc = subprocess.run(cmd)
        metrics = read_last_metrics(trial_dir)
        results.append({
            "trial": i,
            "hparams": hparams,
            "metrics": metrics,
            "returncode": proc.returncode,
            "wallclock_s": round(time.time() - t0, 1),
        })
        with open(os.path.join(args.output, "results.jsonl"), "a") as f:
            f.write(json.dumps(results[-1]) + "\n")

    scored = [r for r in results if metric in r["metrics"] and r["returncode"] == 0]
    if scored:
        best = (max if mode == "max" else min)(scored, key=lambda r: r["metrics"][metric])
        print(f"[sweep] best trial {best['trial']}: {metric}={best['metrics'][metric]}")
        print(json.dumps(best["hparams"], indent=2))
        with open(os.path.join(args.output, "best.json"), "w") as f:
            json.dump(best, f, indent=2)
    else:
        print("[sweep] no trial reported the target metric")


if __name__ == "__main__":
    main()
