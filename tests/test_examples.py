"""Smoke-run the randomwalks examples (the reference's regression pair,
scripts/benchmark.sh:49-50) at tiny settings on CPU."""

import sys

import pytest

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
