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


def _write_quadratic_stub(tmp_path):
    """Stub whose metric is a quadratic of method.x (max at x=0.6)."""
    script = tmp_path / "quad_stub.py"
    script.write_text(
        "import json, os, sys\n"
        "h = json.loads(sys.argv[1])\n"
        "d = h['train.logging_dir']\n"
        "os.makedirs(d, exist_ok=True)\n"
        "x = float(h['method.x'])\n"
        "budget = int(h.get('train.total_steps', 8))\n"
        "score = 1.0 - (x - 0.6) ** 2 + 0.001 * budget\n"
        "with open(os.path.join(d, 'metrics.jsonl'), 'w') as f:\n"
        "    f.write(json.dumps({'reward/mean': score}) + '\\n')\n"
    )
    return script


def test_bayes_sweep(tmp_path):
    from trlx_amd import sweep

    script = _write_quadratic_stub(tmp_path)
    cfg = tmp_path / "space.yml"
    cfg.write_text(yaml.safe_dump({
        "tune_config": {"search_alg": "bayes", "metric": "reward/mean", "mode": "max",
                        "num_samples": 10},
        "method.x": {"strategy": "uniform", "values": [0.0, 1.0]},
    }))
    out = tmp_path / "out"
    sweep.main([str(cfg), str(script), "--output", str(out), "--seed", "1"])
    best = json.load(open(out / "best.json"))
    # GP-EI on a smooth 1-d quadratic should land near the optimum
    assert abs(best["hparams"]["method.x"] - 0.6) < 0.25
    lines = open(out / "results.jsonl").read().strip().splitlines()
    assert len(lines) == 10


def test_hyperband_sweep(tmp_path):
    from trlx_amd import sweep

    script = _write_quadratic_stub(tmp_path)
    cfg = tmp_path / "space.yml"
    cfg.write_text(yaml.safe_dump({
        "tune_config": {"search_alg": "random", "scheduler": "hyperband",
                        "metric": "reward/mean", "mode": "max", "num_samples": 9,
                        "max_budget": 36, "eta": 3,
                        "budget_param": "train.total_steps"},
        "method.x": {"strategy": "uniform", "values": [0.0, 1.0]},
    }))
    out = tmp_path / "out_hb"
    sweep.main([str(cfg), str(script), "--output", str(out), "--seed", "2"])
    results = [json.loads(l) for l in open(out / "results.jsonl")]
    budgets = sorted({r["hparams"]["train.total_steps"] for r in results})
    assert len(budgets) >= 2 and budgets[-1] == 36  # promotions reached max budget
    # later rungs run fewer configs
    n_low = sum(1 for r in results if r["hparams"]["train.total_steps"] == budgets[0])
    n_high = sum(1 for r in results if r["hparams"]["train.total_steps"] == budgets[-1])
    assert n_high < n_low
    assert os.path.exists(out / "best.json")


def test_parallel_sweep(tmp_path):
    from trlx_amd import sweep

    script = _write_quadratic_stub(tmp_path)
    cfg = tmp_path / "space.yml"
    cfg.write_text(yaml.safe_dump({
        "tune_config": {"search_alg": "random", "metric": "reward/mean",
                        "num_samples": 6},
        "method.x": {"strategy": "uniform", "values": [0.0, 1.0]},
    }))
    out = tmp_path / "out_par"
    sweep.main([str(cfg), str(script), "--output", str(out), "--parallel", "3"])
    results = [json.loads(l) for l in open(out / "results.jsonl")]
    assert len(results) == 6 and all(r["returncode"] == 0 for r in results)
