"""Sweep runner tests (parity: reference trlx/sweep.py grammar)."""

import json
import os
import subprocess
import sys

import yaml

from trlx_amd.sweep import generate_trials, parse_space, read_last_metrics, sample_param


def test_grid_trials():
    tune, dims = parse_space({
        "tune_config": {"search_alg": "grid", "metric": "m", "mode": "max"},
        "a.x": {"strategy": "choice", "values": [1, 2]},
        "b.y": {"strategy": "choice", "values": ["p", "q", "r"]},
    })
    trials = list(generate_trials(tune, dims))
    assert len(trials) == 6
    assert {t["a.x"] for t in trials} == {1, 2}
    assert {t["b.y"] for t in trials} == {"p", "q", "r"}


def test_random_trials_strategies():
    tune, dims = parse_space({
        "tune_config": {"search_alg": "random", "num_samples": 12},
        "lr": {"strategy": "loguniform", "values": [1e-5, 1e-1]},
        "bs": {"strategy": "choice", "values": [4, 8]},
        "beta": {"strategy": "uniform", "values": [0.0, 1.0]},
        "steps": {"strategy": "quniform", "values": [10, 100, 10]},
    })
    trials = list(generate_trials(tune, dims, seed=3))
    assert len(trials) == 12
    for t in trials:
        assert 1e-5 <= t["lr"] <= 1e-1
        assert t["bs"] in (4, 8)
        assert 0.0 <= t["beta"] <= 1.0
        assert t["steps"] % 10 == 0
    # deterministic under the same seed
    assert trials == list(generate_trials(tune, dims, seed=3))


def test_read_last_metrics(tmp_path):
    p = tmp_path / "metrics.jsonl"
    p.write_text('{"step": 1, "reward/mean": 0.1}\n{"step": 2, "reward/mean": 0.7}\n')
    m = read_last_metrics(str(tmp_path))
    assert m["reward/mean"] == 0.7


def test_sweep_end_to_end(tmp_path):
    """Two-trial grid sweep over a stub training script."""
    script = tmp_path / "train_stub.py"
    script.write_text(
        "import json, os, sys\n"
        "h = json.loads(sys.argv[1])\n"
        "d = h['train.logging_dir']\n"
        "os.makedirs(d, exist_ok=True)\n"
        "with open(os.path.join(d, 'metrics.jsonl'), 'w') as f:\n"
        "    f.write(json.dumps({'reward/mean': h['method.x'] * 2.0}) + '\\n')\n"
    )
    cfg = tmp_path / "sweep.yml"
    cfg.write_text(yaml.safe_dump({
        "tune_config": {"search_alg": "grid", "metric": "reward/mean", "mode": "max"},
        "method.x": {"strategy": "choice", "values": [1, 3]},
    }))
    out = tmp_path / "results"
    r = subprocess.run([sys.executable, "-m", "trlx_amd.sweep", str(cfg), str(script),
                        "--output", str(out)], capture_output=True, text=True,
                       cwd=os.getcwd())
    assert r.returncode == 0, r.stderr
    lines = [json.loads(l) for l in (out / "results.jsonl").read_text().splitlines()]
    assert len(lines) == 2
    assert "best trial 1" in r.stdout  # x=3 wins under mode=max
