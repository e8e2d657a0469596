"""Benchmark-differ tests (trlx_amd/reference.py, parity: reference
trlx/reference.py W&B branch comparison — offline jsonl version)."""

import json
import os

from trlx_amd.reference import collect_metrics


def test_collect_metrics_bench_and_jsonl(tmp_path):
    # bench-style JSON output
    (tmp_path / "bench_gpt2_ppo.json").write_text(
        json.dumps({"metric": "ppo_samples_per_sec", "value": 123.4}) + "\n")
    # tracker jsonl under a subdir
    sub = tmp_path / "ppo_randomwalks"
    sub.mkdir()
    (sub / "metrics.jsonl").write_text(
        '{"step": 1, "reward/mean": 0.2, "time/forward": 9.0}\n'
        '{"step": 2, "reward/mean": 0.9}\n')
    m = collect_metrics(str(tmp_path))
    assert m["bench_gpt2_ppo.json:ppo_samples_per_sec"] == 123.4
    assert m["ppo_randomwalks:reward/mean"] == 0.9
    # time/* and step keys are excluded from the comparison table
    assert not any("time/forward" in k or k.endswith(":step") for k in m)
