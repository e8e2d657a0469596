#!/bin/bash
# Pinned regression suite (parity: reference scripts/benchmark.sh:49-68).
# Fast pair: randomwalks ILQL + PPO (tiny, no network); then the flagship
# GPT-2 PPO bench.  Results land in $BENCH_OUTPUT_DIR (default: bench_out/).
set -e
export PYTHONPATH="$(pwd)${PYTHONPATH:+:$PYTHONPATH}"
OUT="${BENCH_OUTPUT_DIR:-bench_out}"
mkdir -p "$OUT"

CONF_RW='{"train.total_steps": 8, "train.epochs": 2, "train.batch_size": 32,
          "train.eval_interval": 8, "train.checkpoint_interval": 1000,
          "train.tracker": "jsonl", "train.save_best": false}'

echo "[benchmark] randomwalks ILQL"
python examples/randomwalks/ilql_randomwalks.py \
  "$(echo $CONF_RW | python -c 'import json,sys; d=json.load(sys.stdin); d["train.logging_dir"]="'"$OUT"'/ilql_randomwalks"; print(json.dumps(d))')"

echo "[benchmark] randomwalks PPO"
python examples/randomwalks/ppo_randomwalks.py \
  "$(echo $CONF_RW | python -c 'import json,sys; d=json.load(sys.stdin); d["train.logging_dir"]="'"$OUT"'/ppo_randomwalks"; d["method.num_rollouts"]=32; d["method.chunk_size"]=32; print(json.dumps(d))')"

echo "[benchmark] flagship GPT-2 PPO samples/sec"
python bench.py --steps 2 --warmup 1 > "$OUT/bench_gpt2_ppo.json"
cat "$OUT/bench_gpt2_ppo.json"
