"""Smoke-run the randomwalks examples (the reference's regression pair,
scripts/benchmark.sh:49-50) at tiny settings on CPU."""

import sys

import pytest
import torch

sys.path.insert(0, ".")


def test_randomwalks_task_properties():
    from examples.randomwalks import generate_random_walks

    metric_fn, eval_prompts, walks, logit_mask = generate_random_walks(seed=1002, n_walks=50)
    assert len(walks) == 50
    assert all(w[-1] == "a" or len(w) == 10 for w in walks)
    scores = metric_fn(walks)
    assert len(scores["optimality"]) == 50
    assert all(0 <= s <= 1 for s in scores["optimality"])
    # the sampled walks themselves are valid paths: never invalid length 100
    assert all(l <= 10 or l == 100 for l in scores["lengths"])
    assert logit_mask.shape == (21, 21)


@pytest.mark.parametrize("which", ["ppo", "ilql"])
def test_randomwalks_examples_run(which, tmp_path):
    from examples.randomwalks import ilql_randomwalks, ppo_randomwalks

    overrides = {
        "train.total_steps": 2,
        "train.epochs": 1,
        "train.batch_size": 8,
        "train.eval_interval": 2,
        "train.checkpoint_interval": 100,
        "train.checkpoint_dir": str(tmp_path / "ckpt"),
        "train.tracker": None,
        "train.save_best": False,
    }
    if which == "ppo":
        overrides.update({"method.num_rollouts": 8, "method.chunk_size": 8,
                          "method.ppo_epochs": 1, "method.gen_kwargs": dict(
                              max_new_tokens=4, top_k=0, top_p=1.0, do_sample=True)})
        ppo_randomwalks.main(overrides)
    else:
        overrides.update({"method.gen_kwargs": dict(max_new_tokens=4, top_k=5, beta=1,
                                                    temperature=1.0)})
        ilql_randomwalks.main(overrides)


@pytest.mark.parametrize("which", ["ppo", "ilql", "sft", "rft", "ppo_dense"])
def test_sentiments_examples_run(which, tmp_path, monkeypatch):
    monkeypatch.syspath_prepend("examples")
    import importlib

    mod = importlib.import_module({
        "ppo": "ppo_sentiments", "ilql": "ilql_sentiments", "sft": "sft_sentiments",
        "rft": "rft_sentiments", "ppo_dense": "ppo_dense_sentiments",
    }[which])
    overrides = {
        "train.total_steps": 2,
        "train.epochs": 1,
        "train.batch_size": 4,
        "train.eval_interval": 2,
        "train.checkpoint_interval": 100,
        "train.checkpoint_dir": str(tmp_path / "ckpt"),
        "train.tracker": None,
        "train.save_best": False,
        "train.seq_length": 32,
        "model.model_extra_configs": {"config": __import__("conftest").tiny_config().to_dict()},
    }
    if which in ("ppo", "ppo_dense"):
        overrides.update({"method.num_rollouts": 4, "method.chunk_size": 4,
                          "method.ppo_epochs": 1,
                          "method.gen_kwargs": dict(max_new_tokens=4, top_k=0, top_p=1.0, do_sample=True)})
    elif which == "rft":
        overrides.update({"method.n_generations_per_prompt": 2,
                          "method.gen_kwargs": dict(max_new_tokens=4, do_sample=True)})
    elif which == "ilql":
        overrides.update({"method.gen_kwargs": dict(max_new_tokens=4, top_k=5, beta=1, temperature=1.0)})
    mod.main(overrides)


def test_summarize_rlhf_pipeline(tmp_path, monkeypatch):
    """Three-stage RLHF (SFT -> reward model -> PPO) end-to-end at tiny scale
    (parity: reference examples/summarize_rlhf/*)."""
    monkeypatch.syspath_prepend("examples/summarize_rlhf")
    monkeypatch.setenv("TRLX_AMD_SUMMARIZE_DIR", str(tmp_path))
    import importlib

    import conftest

    tiny = conftest.tiny_config()

    # stage 1: SFT
    sft = importlib.import_module("sft_train")
    importlib.reload(sft)  # re-read TRLX_AMD_SUMMARIZE_DIR
    hf_dir = sft.main({
        "train.total_steps": 2, "train.epochs": 1, "train.batch_size": 8,
        "train.eval_interval": 2, "train.checkpoint_interval": 100,
        "train.tracker": None, "train.save_best": False, "train.seq_length": 32,
        "model.model_extra_configs": {"config": tiny.to_dict()},
    })
    assert (tmp_path / "sft" / "hf_model" / "config.json").exists()

    # stage 2: reward model (trunk initialized from the SFT checkpoint)
    rm_mod = importlib.import_module("train_reward_model")
    importlib.reload(rm_mod)
    rm_dir, acc = rm_mod.main(sft_dir=hf_dir, n_pairs=32, epochs=1, batch_size=8,
                              seq_length=32)
    assert (tmp_path / "rm" / "rm_model.pt").exists()
    assert 0.0 <= acc <= 1.0

    # stage 3: PPO scored by the trained RM
    ppo = importlib.import_module("ppo_summarize")
    importlib.reload(ppo)
    ppo.main({
        "train.total_steps": 2, "train.epochs": 1, "train.batch_size": 4,
        "train.eval_interval": 2, "train.checkpoint_interval": 100,
        "train.tracker": None, "train.save_best": False, "train.seq_length": 32,
        "method.num_rollouts": 4, "method.chunk_size": 4, "method.ppo_epochs": 1,
        "method.gen_kwargs": dict(max_new_tokens=4, top_k=0, top_p=1.0, do_sample=True),
    }, sft_dir=hf_dir, rm_dir=rm_dir)


def test_reward_model_pairwise_loss_semantics():
    """Vectorized pairwise loss matches the reference's per-row definition
    (reference reward_model.py:61-90) on a hand-checkable case."""
    import importlib
    import sys as _sys

    if "examples/summarize_rlhf" not in _sys.path:
        _sys.path.insert(0, "examples/summarize_rlhf")
    rm_lib = importlib.import_module("reward_model")
    import conftest

    model = rm_lib.RewardModel.from_pretrained(conftest.tiny_config(), pad_token_id=2)
    # chosen/rejected share a 3-token prefix, diverge at position 3; pads at 6
    chosen = torch.tensor([[5, 6, 7, 10, 11, 1, 2, 2]])
    rejected = torch.tensor([[5, 6, 7, 20, 21, 1, 2, 2]])
    ids = torch.cat([chosen, rejected])
    out = model(ids)
    rewards = model._rewards(ids)
    # reference loop: loss over positions div..end (3..6)
    diff = rewards[0, 3:6] - rewards[1, 3:6]
    expect = -torch.log(torch.sigmoid(diff)).mean()
    assert torch.allclose(out["loss"], expect, atol=1e-6)
    assert torch.allclose(out["chosen_end_scores"], rewards[0, 5])
    # inference path: identical halves -> only chosen_end_scores
    out_inf = model(torch.cat([chosen, chosen]))
    assert set(out_inf.keys()) == {"chosen_end_scores"}


def test_hh_example_and_reward_server(tmp_path, monkeypatch):
    """HH PPO example + the out-of-band reward server (parity: reference
    examples/hh/ppo_hh.py + Triton channel)."""
    monkeypatch.syspath_prepend("examples/hh")
    import importlib

    import conftest

    hh_task = importlib.import_module("hh_task")
    scores = hh_task.oracle_reward(["\n\nHuman: hi\n\nAssistant: sure I can help"])
    assert scores[0] > 0

    # reward server endpoint (in-process TestClient; oracle mode)
    from fastapi.testclient import TestClient

    server = importlib.import_module("reward_server")
    client = TestClient(server.app)
    assert client.get("/health").json()["ok"]
    r = client.post("/reward", json={"samples": ["\n\nHuman: q\n\nAssistant: go away"]})
    assert r.status_code == 200 and r.json()["scores"][0] < 0

    # tiny PPO run with the in-process oracle
    ppo_hh = importlib.import_module("ppo_hh")
    ppo_hh.main({
        "train.total_steps": 2, "train.epochs": 1, "train.batch_size": 4,
        "train.eval_interval": 2, "train.checkpoint_interval": 100,
        "train.checkpoint_dir": str(tmp_path / "ckpt"), "train.tracker": None,
        "train.save_best": False, "train.seq_length": 48,
        "model.model_extra_configs": {"config": conftest.tiny_config().to_dict()},
        "model.num_layers_unfrozen": 1,
        "method.num_rollouts": 4, "method.chunk_size": 4, "method.ppo_epochs": 1,
        "method.gen_kwargs": dict(max_new_tokens=4, top_k=0, top_p=1.0, do_sample=True),
    })


def test_reward_server_with_trained_rm(tmp_path, monkeypatch):
    """reward_server serves a trained RM checkpoint (not just the oracle)."""
    monkeypatch.syspath_prepend("examples/hh")
    monkeypatch.syspath_prepend("examples/summarize_rlhf")
    import importlib

    import conftest
    from fastapi.testclient import TestClient

    rm_lib = importlib.import_module("reward_model")
    model = rm_lib.RewardModel.from_pretrained(conftest.tiny_config(), pad_token_id=2)
    model.save_checkpoint(str(tmp_path / "rm"))

    server = importlib.import_module("reward_server")
    server._rm["model"] = None
    server.load_rm(str(tmp_path / "rm"))
    try:
        client = TestClient(server.app)
        assert client.get("/health").json()["rm"] is True
        r = client.post("/reward", json={"samples": ["hello there", "goodbye"]})
        assert r.status_code == 200
        scores = r.json()["scores"]
        assert len(scores) == 2 and all(isinstance(x, float) for x in scores)
    finally:
        server._rm["model"] = None
        server._rm["tok"] = None


@pytest.mark.parametrize("which", ["architext", "simulacra", "llama", "peft"])
def test_examples_tail_run(which, tmp_path, monkeypatch):
    """Smoke-run the round-2 example tail (architext, simulacra, llama/peft
    sentiment variants) at tiny settings on CPU."""
    monkeypatch.syspath_prepend("examples")
    import importlib

    mod = importlib.import_module({
        "architext": "architext", "simulacra": "simulacra",
        "llama": "ppo_sentiments_llama", "peft": "ppo_sentiments_peft",
    }[which])
    overrides = {
        "train.total_steps": 2,
        "train.epochs": 1,
        "train.batch_size": 4,
        "train.eval_interval": 2,
        "train.checkpoint_interval": 100,
        "train.checkpoint_dir": str(tmp_path / "ckpt"),
        "train.tracker": None,
        "train.save_best": False,
        "train.seq_length": 32,
    }
    if which in ("architext", "peft"):
        overrides["model.model_extra_configs"] = {
            "config": __import__("conftest").tiny_config().to_dict()}
    if which == "llama":
        overrides["model.model_extra_configs"] = {
            "config": __import__("conftest").tiny_config(
                arch_name="llama", norm="rmsnorm", position_encoding="rope",
                activation="silu", swiglu=True, attn_bias=False, mlp_bias=False,
                intermediate_size=128, tie_word_embeddings=False).to_dict()}
    if which in ("architext", "llama", "peft"):
        overrides.update({"method.num_rollouts": 4, "method.chunk_size": 4,
                          "method.ppo_epochs": 1,
                          "method.gen_kwargs": dict(max_new_tokens=4, top_k=0, top_p=1.0,
                                                    do_sample=True)})
    else:  # simulacra (ILQL)
        overrides.update({"method.gen_kwargs": dict(max_new_tokens=4, top_k=5, beta=1,
                                                    temperature=1.0)})
    mod.main(overrides)


def test_grounded_program_synthesis_run(tmp_path, monkeypatch):
    monkeypatch.syspath_prepend("examples/grounded_program_synthesis")
    import importlib

    lang = importlib.import_module("lang")
    # grounded grading: gold programs score +1, junk scores -1
    data = lang.make_dataset(8, seed=3)
    assert lang.reward_fn([f"{p} {g}" for p, g in data]) == [1.0] * 8
    assert lang.reward_fn(["nonsense"]) == [-1.0]

    mod = importlib.import_module("train_trlx")
    overrides = {
        "train.total_steps": 2,
        "train.epochs": 1,
        "train.batch_size": 4,
        "train.eval_interval": 2,
        "train.checkpoint_interval": 100,
        "train.checkpoint_dir": str(tmp_path / "ckpt"),
        "train.tracker": None,
        "train.save_best": False,
        "train.seq_length": 48,
        "model.model_extra_configs": {"config": __import__("conftest").tiny_config().to_dict()},
        "method.num_rollouts": 4,
        "method.chunk_size": 4,
        "method.ppo_epochs": 1,
        "method.gen_kwargs": dict(max_new_tokens=4, top_k=0, top_p=1.0, do_sample=True),
    }
    mod.main(overrides)


def test_alpaca_sft_example(tmp_path):
    import os
    import subprocess
    import sys

    REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    env = dict(os.environ)
    r = subprocess.run(
        [sys.executable, "examples/alpaca/sft_alpaca.py",
         "--hparams", '{"train.total_steps": 2, "train.eval_interval": 100, '
                      '"train.checkpoint_interval": 1000000000, '
                      '"train.checkpoint_dir": "%s"}' % tmp_path],
        cwd=REPO, env=env, capture_output=True, text=True, timeout=600)
    assert r.returncode == 0, r.stderr[-2000:]


def test_t5_translation_and_ilql_examples(tmp_path):
    import os
    import subprocess
    import sys

    REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    for script, hp in [
        ("examples/ppo_translation_t5.py",
         '{"train.total_steps": 2, "method.num_rollouts": 8, "method.chunk_size": 8, '
         '"train.eval_interval": 100, "train.checkpoint_interval": 1000000000, '
         '"train.checkpoint_dir": "%s"}' % tmp_path),
        ("examples/ilql_sentiments_t5.py",
         '{"train.total_steps": 2, "train.eval_interval": 100, '
         '"train.checkpoint_interval": 1000000000, '
         '"train.checkpoint_dir": "%s"}' % tmp_path),
    ]:
        r = subprocess.run([sys.executable, script, hp], cwd=REPO, env=dict(os.environ),
                           capture_output=True, text=True, timeout=600)
        assert r.returncode == 0, (script, r.stderr[-2000:])


def test_inference_checkpoint_example(tmp_path):
    import os
    import subprocess
    import sys

    import torch

    REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    from conftest import tiny_config
    from trlx_amd.models.modeling_ppo import AutoModelForCausalLMWithValueHead

    torch.manual_seed(0)
    m = AutoModelForCausalLMWithValueHead.from_config(tiny_config())
    m.save_pretrained(str(tmp_path / "ckpt"))
    r = subprocess.run(
        [sys.executable, "examples/inference_checkpoint.py", "--model",
         str(tmp_path / "ckpt"), "--tokenizer", "byte", "--max-new-tokens", "6",
         "--greedy", "hello world"],
        cwd=REPO, env=dict(os.environ), capture_output=True, text=True, timeout=300)
    assert r.returncode == 0, r.stderr[-2000:]
    assert "=== 'hello world'" in r.stdout
