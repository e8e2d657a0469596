"""Driver-contract dry-run of bench.py: the exact torchrun launch shape the
driver uses for multi-GPU scaling (one rank per device, RANK/WORLD_SIZE from
the env, MAX-over-ranks aggregation, ONE JSON line from rank 0) — exercised
here on CPU/gloo with a tiny model so first contact with a real 8-GPU node
is a solved problem."""

import json
import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

TINY = ["--tiny-smoke", "--no-secondary", "--num-rollouts", "8", "--chunk-size", "4",
        "--batch-size", "4", "--seq-len", "64", "--prompt-len", "8",
        "--max-new-tokens", "4", "--num-prompts", "16"]


def _last_json_line(stdout: str) -> dict:
    lines = [l for l in stdout.strip().splitlines() if l.startswith("{")]
    assert lines, f"no JSON line in output:\n{stdout}"
    return json.loads(lines[-1])


@pytest.mark.parametrize("world", [2, 4, 8])
def test_torchrun_bench_contract(world, tmp_path):
    env = dict(os.environ)
    env["HIP_VISIBLE_DEVICES"] = ""
    env["CUDA_VISIBLE_DEVICES"] = ""
    port = str(29650 + world)
    cmd = [
        sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
        f"--nproc-per-node={world}", "--master-addr", "127.0.0.1", "--master-port", port,
        os.path.join(REPO, "bench.py"), "--gpus", str(world), "--steps", "1",
        "--warmup", "0", *TINY,
    ]
    out = subprocess.run(cmd, capture_output=True, text=True, timeout=600, cwd=REPO, env=env)
    assert out.returncode == 0, out.stderr[-4000:]
    result = _last_json_line(out.stdout)
    assert result["n_gpus"] == world
    assert result["value"] > 0
    assert result["config"]["parallelism"] == f"dp{world}"
    assert result["config"]["global_batch"] == 4 * world
    assert result["metric"] == "ppo_samples_per_sec"
    assert "phases_ms_per_step" in result
    # the preflight must have verified the collectives before the run
    assert "[comm.preflight] OK" in out.stderr


def test_bench_single_process_contract():
    env = dict(os.environ)
    env["HIP_VISIBLE_DEVICES"] = ""
    env["CUDA_VISIBLE_DEVICES"] = ""
    for k in ("RANK", "WORLD_SIZE", "LOCAL_RANK", "MASTER_ADDR", "MASTER_PORT"):
        env.pop(k, None)
    cmd = [sys.executable, os.path.join(REPO, "bench.py"), "--gpus", "1", "--steps", "1",
           "--warmup", "0", *TINY]
    out = subprocess.run(cmd, capture_output=True, text=True, timeout=600, cwd=REPO, env=env)
    assert out.returncode == 0, out.stderr[-4000:]
    result = _last_json_line(out.stdout)
    assert result["n_gpus"] == 1 and result["value"] > 0
