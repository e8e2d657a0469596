"""Benchmark differ: ``python -m trlx_amd.reference <branch> --against <branch>``.

Parity target: reference trlx/reference.py (W&B report comparing benchmark
runs of two git branches).  Offline version: runs ``scripts/benchmark.sh``
for each branch if its results are missing (keyed by a content hash of the
tree), then prints a side-by-side metric comparison table from the jsonl
trackers/bench outputs.
"""

import argparse
import hashlib
import json
import os
import subprocess
import sys


def tree_hash(ref: str) -> str:
    """Content hash of a git ref's tree (reference hashes branch trees too)."""
    out = subprocess.check_output(["git", "rev-parse", f"{ref}^{{tree}}"])
    return hashlib.sha1(out.strip()).hexdigest()[:12]


def run_benchmark(ref: str, out_dir: str) -> str:
    """Check out ``ref`` into a temp worktree and run scripts/benchmark.sh."""
    key = tree_hash(ref)
    result_dir = os.path.join(out_dir, key)
    marker = os.path.join(result_dir, "DONE")
    if os.path.exists(marker):
        print(f"[reference] cached results for {ref} ({key})")
        return result_dir
    os.makedirs(result_dir, exist_ok=True)
    worktree = os.path.join(out_dir, f"tree_{key}")
    if not os.path.exists(worktree):
        subprocess.run(["git", "worktree", "add", "--detach", worktree, ref], check=True)
    env = dict(os.environ, BENCH_OUTPUT_DIR=os.path.abspath(result_dir))
    subprocess.run(["bash", "scripts/benchmark.sh"], cwd=worktree, env=env, check=True)
    with open(marker, "w") as f:
        f.write(ref)
    return result_dir


def collect_metrics(result_dir: str) -> dict:
    metrics = {}
    for root, _dirs, files in os.walk(result_dir):
        for fn in files:
            if fn.endswith(".json") and fn.startswith("bench"):
                with open(os.path.join(root, fn)) as f:
                    try:
                        rec = json.loads(f.read().strip().splitlines()[-1])
                        metrics[f"{fn}:{rec.get('metric', '?')}"] = rec.get("value")
                    except (json.JSONDecodeError, IndexError):
                        continue
            if fn == "metrics.jsonl":
                last = {}
                with open(os.path.join(root, fn)) as f:
                    for line in f:
                        try:
                            last.update(json.loads(line))
                        except json.JSONDecodeError:
                            continue
                tag = os.path.relpath(root, result_dir)
                for k, v in last.items():
                    if isinstance(v, (int, float)) and not k.startswith(("time", "step")):
                        metrics[f"{tag}:{k}"] = v
    return metrics


def main():
    parser = argparse.ArgumentParser(description="compare benchmark runs of two branches")
    parser.add_argument("branch", help="branch/ref to benchmark")
    parser.add_argument("--against", default="main", help="baseline branch/ref")
    parser.add_argument("--out", default="benchmark_runs")
    args = parser.parse_args()

    os.makedirs(args.out, exist_ok=True)
    base_dir = run_benchmark(args.against, args.out)
    new_dir = run_benchmark(args.branch, args.out)

    base = collect_metrics(base_dir)
    new = collect_metrics(new_dir)
    keys = sorted(set(base) | set(new))
    width = max((len(k) for k in keys), default=10)
    print(f"\n{'metric':<{width}}  {args.against:>14}  {args.branch:>14}  {'delta':>10}")
    for k in keys:
        b = base.get(k)
        n = new.get(k)
        delta = ""
        if isinstance(b, (int, float)) and isinstance(n, (int, float)) and b:
            delta = f"{100 * (n - b) / abs(b):+.1f}%"
        print(f"{k:<{width}}  {str(b):>14}  {str(n):>14}  {delta:>10}")


if __name__ == "__main__":
    main()
