"""Hyperparameter sweep runner: ``python -m trlx_amd.sweep config.yml script.py``.

Parity target: reference trlx/sweep.py (Ray Tune search algorithms +
schedulers + W&B reports).  Ray and W&B are not available offline, so the
same YAML search-space grammar runs on a local subprocess executor with:

- search_alg: grid | random | bayes   (bayes = Gaussian-process expected
  improvement on the encoded space, scikit-learn — the reference's BayesOpt
  role, trlx/sweep.py:103-133)
- scheduler: hyperband                 (ASHA-style successive halving over a
  budget parameter, default train.total_steps — the reference's HyperBand,
  trlx/sweep.py:136-158)
- --parallel N                         (N concurrent trials; single-GPU
  trials are pinned round-robin to visible devices)

Results land in <output>/results.jsonl + best.json.

Search-space grammar (same keys as the reference's YAML):
  tune_config:
    mode: max | min
    metric: reward/mean
    search_alg: grid | random | bayes
    num_samples: 16
    scheduler: hyperband          # optional
    budget_param: train.total_steps
    max_budget: 64
    eta: 3
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
import math
import os
import random
import subprocess
import sys
import time
from concurrent.futures import ThreadPoolExecutor
from typing import Any, Dict, List, Optional


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
    else:  # random (bayes adds its own proposal loop on top)
        rng = random.Random(seed)
        n = int(tune_config.get("num_samples", 8))
        for _ in range(n):
            yield {p: sample_param(s, v, rng) for p, (s, v) in dims.items()}


# ---- encoded space for the GP (bayes) ---------------------------------------


def _encode(dims: Dict, hparams: Dict) -> List[float]:
    """Map hparams onto [0, 1]^d (log scale for loguniform; choice -> index)."""
    x = []
    for p, (s, v) in dims.items():
        val = hparams[p]
        if s == "choice":
            x.append(v.index(val) / max(len(v) - 1, 1))
        elif s == "uniform":
            x.append((val - v[0]) / (v[1] - v[0]))
        elif s == "loguniform":
            x.append((math.log(val) - math.log(v[0])) / (math.log(v[1]) - math.log(v[0])))
        elif s == "quniform":
            x.append((val - v[0]) / (v[1] - v[0]))
    return x


def _decode(dims: Dict, x: List[float]) -> Dict:
    out = {}
    for (p, (s, v)), xi in zip(dims.items(), x):
        xi = min(max(xi, 0.0), 1.0)
        if s == "choice":
            out[p] = v[round(xi * (len(v) - 1))]
        elif s == "uniform":
            out[p] = v[0] + xi * (v[1] - v[0])
        elif s == "loguniform":
            out[p] = math.exp(math.log(v[0]) + xi * (math.log(v[1]) - math.log(v[0])))
        elif s == "quniform":
            lo, hi, q = v
            out[p] = round((lo + xi * (hi - lo)) / q) * q
    return out


def propose_bayes(dims: Dict, observed: List, mode: str, rng: random.Random,
                  n_candidates: int = 512) -> Dict:
    """GP expected-improvement proposal over the encoded space; pending trials
    enter with the observed mean ("constant liar") so parallel batches spread
    out.  Falls back to random until 2 scored observations exist."""
    scored = [(x, y) for x, y in observed if y is not None]
    if len(scored) < 2:
        return {p: sample_param(s, v, rng) for p, (s, v) in dims.items()}
    liar = sum(y for _, y in scored) / len(scored)
    X = [x for x, _ in observed]
    y = [(yv if yv is not None else liar) for _, yv in observed]
    sign = 1.0 if mode == "max" else -1.0
    y = [sign * v for v in y]

    import numpy as np
    from sklearn.gaussian_process import GaussianProcessRegressor
    from sklearn.gaussian_process.kernels import RBF, ConstantKernel, WhiteKernel

    kernel = ConstantKernel(1.0) * RBF(length_scale=0.25) + WhiteKernel(1e-3)
    gp = GaussianProcessRegressor(kernel=kernel, normalize_y=True)
    gp.fit(np.asarray(X), np.asarray(y))
    best = max(y)
    cand = np.random.RandomState(rng.randrange(2**31)).rand(n_candidates, len(dims))
    mu, sigma = gp.predict(cand, return_std=True)
    sigma = np.maximum(sigma, 1e-9)
    z = (mu - best) / sigma
    from scipy.stats import norm

    ei = (mu - best) * norm.cdf(z) + sigma * norm.pdf(z)
    return _decode(dims, list(cand[int(np.argmax(ei))]))


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


class Executor:
    """Runs trials as subprocesses, optionally in parallel with round-robin
    single-GPU pinning (the reference parallelizes through Ray workers)."""

    def __init__(self, args, metric: str):
        self.args = args
        self.metric = metric
        self.counter = 0
        try:
            import torch

            self.n_devices = torch.cuda.device_count() if torch.cuda.is_available() else 0
        except Exception:
            self.n_devices = 0

    def run(self, idx: int, hparams: Dict) -> Dict:
        args = self.args
        trial_dir = os.path.join(args.output, f"trial_{idx:03d}")
        hparams = dict(hparams)
        hparams["train.logging_dir"] = trial_dir
        hparams["train.tracker"] = "jsonl"
        hparams["train.checkpoint_dir"] = os.path.join(trial_dir, "ckpts")
        print(f"[sweep] trial {idx}: {hparams}")
        env = dict(os.environ)
        if args.num_gpus > 1:
            cmd = [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
                   f"--nproc-per-node={args.num_gpus}", "--master-addr", "127.0.0.1",
                   "--master-port", str(29400 + idx % 100),
                   args.script, json.dumps(hparams)]
        else:
            cmd = [sys.executable, args.script, json.dumps(hparams)]
            if args.parallel > 1 and self.n_devices > 1:
                env["HIP_VISIBLE_DEVICES"] = str(idx % self.n_devices)
                env["CUDA_VISIBLE_DEVICES"] = str(idx % self.n_devices)
        t0 = time.time()
        proc = subprocess.run(cmd, env=env)
        metrics = read_last_metrics(trial_dir)
        result = {
            "trial": idx,
            "hparams": hparams,
            "metrics": metrics,
            "returncode": proc.returncode,
            "wallclock_s": round(time.time() - t0, 1),
        }
        with open(os.path.join(args.output, "results.jsonl"), "a") as f:
            f.write(json.dumps(result) + "\n")
        return result

    def score(self, result: Dict) -> Optional[float]:
        if result["returncode"] != 0:
            return None
        v = result["metrics"].get(self.metric)
        return float(v) if v is not None else None


def run_hyperband(ex: Executor, tune_config: Dict, dims: Dict, seed: int, mode: str,
                  parallel: int) -> List[Dict]:
    """ASHA-style successive halving: num_samples random configs start at
    max_budget / eta^(rungs-1); the top 1/eta of each rung re-run with eta x
    the budget (reference HyperBand role, trlx/sweep.py:136-158)."""
    budget_param = tune_config.get("budget_param", "train.total_steps")
    max_budget = int(tune_config.get("max_budget", 64))
    eta = int(tune_config.get("eta", 3))
    n = int(tune_config.get("num_samples", 9))
    rungs = max(1, int(math.log(n, eta)) + 1)
    b0 = max(1, max_budget // (eta ** (rungs - 1)))

    rng = random.Random(seed)
    configs = [{p: sample_param(s, v, rng) for p, (s, v) in dims.items()} for _ in range(n)]
    results = []
    idx = 0
    budget = b0
    sign = 1.0 if mode == "max" else -1.0
    for rung in range(rungs):
        print(f"[sweep] hyperband rung {rung}: {len(configs)} configs at "
              f"{budget_param}={budget}")
        batch = []
        for cfg in configs:
            h = dict(cfg)
            h[budget_param] = budget
            batch.append((idx, h))
            idx += 1
        with ThreadPoolExecutor(max_workers=max(parallel, 1)) as pool:
            rung_results = list(pool.map(lambda t: ex.run(*t), batch))
        results.extend(rung_results)
        scored = [(r, ex.score(r)) for r in rung_results]
        scored = [(r, s) for r, s in scored if s is not None]
        if not scored or rung == rungs - 1:
            break
        scored.sort(key=lambda rs: -sign * rs[1])
        keep = max(1, len(scored) // eta)
        configs = [{p: r["hparams"][p] for p in dims} for r, _ in scored[:keep]]
        budget = min(budget * eta, max_budget)
    return results


def run_bayes(ex: Executor, tune_config: Dict, dims: Dict, seed: int, mode: str,
              parallel: int) -> List[Dict]:
    n = int(tune_config.get("num_samples", 16))
    rng = random.Random(seed)
    observed = []  # (encoded_x, score_or_None)
    results = []
    idx = 0
    while idx < n:
        batch = []
        for _ in range(min(max(parallel, 1), n - idx)):
            h = propose_bayes(dims, observed, mode, rng)
            observed.append([_encode(dims, h), None])  # pending (constant liar)
            batch.append((idx, h, len(observed) - 1))
            idx += 1
        with ThreadPoolExecutor(max_workers=max(parallel, 1)) as pool:
            outs = list(pool.map(lambda t: (t[2], ex.run(t[0], t[1])), batch))
        for slot, r in outs:
            observed[slot][1] = ex.score(r)
            results.append(r)
    return results


def main(argv=None):
    parser = argparse.ArgumentParser(description="trlx_amd hyperparameter sweep")
    parser.add_argument("config", help="sweep YAML (search space + tune_config)")
    parser.add_argument("script", help="training script taking JSON hparams as argv[1]")
    parser.add_argument("--num-gpus", type=int, default=1, help="GPUs per trial")
    parser.add_argument("--parallel", type=int, default=1, help="concurrent trials")
    parser.add_argument("--output", default="sweep_results", help="results directory")
    parser.add_argument("--seed", type=int, default=0)
    args = parser.parse_args(argv)

    import yaml

    with open(args.config) as f:
        space = yaml.safe_load(f)
    tune_config, dims = parse_space(space)
    metric = tune_config.get("metric", "reward/mean")
    mode = tune_config.get("mode", "max")

    os.makedirs(args.output, exist_ok=True)
    ex = Executor(args, metric)

    alg = tune_config.get("search_alg", "grid")
    scheduler = tune_config.get("scheduler")
    if scheduler == "hyperband":
        results = run_hyperband(ex, tune_config, dims, args.seed, mode, args.parallel)
    elif alg == "bayes":
        results = run_bayes(ex, tune_config, dims, args.seed, mode, args.parallel)
    else:
        trials = list(enumerate(generate_trials(tune_config, dims, args.seed)))
        with ThreadPoolExecutor(max_workers=max(args.parallel, 1)) as pool:
            results = list(pool.map(lambda t: ex.run(*t), trials))

    scored = [(r, ex.score(r)) for r in results]
    scored = [(r, s) for r, s in scored if s is not None]
    if scored:
        best, best_score = (max if mode == "max" else min)(scored, key=lambda rs: rs[1])
        print(f"[sweep] best trial {best['trial']}: {metric}={best_score}")
        print(json.dumps(best["hparams"], indent=2))
        with open(os.path.join(args.output, "best.json"), "w") as f:
            json.dump(best, f, indent=2)
    else:
        print("[sweep] no trial reported the target metric")


if __name__ == "__main__":
    main()
