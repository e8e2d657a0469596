"""End-to-end trainer tests on CPU (parity: reference tests/test_trainers.py —
tiny runs, checkpoint layout incl. best_checkpoint, grad-accumulation)."""

import os

import pytest
import torch

import trlx_amd
from trlx_amd.data.default_configs import (
    default_ilql_config,
    default_ppo_config,
    default_sft_config,
)
from trlx_amd.models.nn.config import TransformerConfig

from conftest import tiny_config


def _tiny_model_cfg(cfg, tmp_path, trainer_name=None, **train_overrides):
    tiny = tiny_config()
    cfg.model.model_path = "tiny"
    cfg.model.model_extra_configs = {"config": tiny.to_dict()}
    cfg.tokenizer.tokenizer_path = "byte"
    cfg.train.seq_length = 48
    cfg.train.batch_size = 4
    cfg.train.total_steps = 2
    cfg.train.eval_interval = 2
    cfg.train.checkpoint_interval = 2
    cfg.train.tracker = None
    cfg.train.checkpoint_dir = str(tmp_path / "ckpts")
    cfg.method.gen_kwargs = dict(max_new_tokens=6, top_k=0, top_p=1.0, do_sample=True)
    for k, v in train_overrides.items():
        setattr(cfg.train, k, v)
    return cfg


def test_ppo_end_to_end(tmp_path):
    cfg = _tiny_model_cfg(default_ppo_config(), tmp_path)
    cfg.model.num_layers_unfrozen = 1
    cfg.method.num_rollouts = 8
    cfg.method.chunk_size = 4
    cfg.method.ppo_epochs = 2

    def reward_fn(samples, prompts, outputs, **kw):
        return [float(len(o)) for o in outputs]

    trainer = trlx_amd.train(
        reward_fn=reward_fn,
        prompts=["hello", "world", "foo bar", "baz"] * 2,
        eval_prompts=["hello", "sky"] * 2,
        config=cfg,
    )
    assert trainer.iter_count == 2
    # checkpoint layout: checkpoint_XX + best_checkpoint + hf_model inside
    ckpts = os.listdir(cfg.train.checkpoint_dir)
    assert any(c.startswith("checkpoint_") for c in ckpts)
    ck = [c for c in ckpts if c.startswith("checkpoint_")][0]
    assert os.path.exists(os.path.join(cfg.train.checkpoint_dir, ck, "hf_model", "config.json"))
    assert os.path.exists(os.path.join(cfg.train.checkpoint_dir, ck, "state.pt"))
    assert "best_checkpoint" in ckpts


def test_ppo_dense_rewards(tmp_path):
    """Per-token rewards path (reference examples/ppo_dense_sentiments.py)."""
    cfg = _tiny_model_cfg(default_ppo_config(), tmp_path)
    cfg.model.num_layers_unfrozen = -1
    cfg.method.num_rollouts = 4
    cfg.method.chunk_size = 4
    cfg.method.ppo_epochs = 1
    cfg.train.save_best = False

    def reward_fn(samples, prompts, outputs, **kw):
        return [[0.1] * max(len(o), 1) for o in outputs]

    trainer = trlx_amd.train(
        reward_fn=reward_fn,
        prompts=["a", "bb", "ccc", "dddd"],
        eval_prompts=["a", "bb", "ccc", "dddd"],
        config=cfg,
    )
    assert trainer.iter_count == 2


def test_ilql_end_to_end(tmp_path):
    cfg = _tiny_model_cfg(default_ilql_config(), tmp_path)
    cfg.train.save_best = False
    cfg.method.gen_kwargs = dict(max_new_tokens=6, top_k=5, beta=1, temperature=1.0)

    trainer = trlx_amd.train(
        samples=["hello world", "goodbye world", "lorem", "ipsum"],
        rewards=[1.0, -1.0, 0.5, -0.5],
        eval_prompts=["hello", "good"] * 2,
        config=cfg,
    )
    assert trainer.iter_count == 2


def test_sft_end_to_end(tmp_path):
    cfg = _tiny_model_cfg(default_sft_config(), tmp_path)
    cfg.train.save_best = False
    trainer = trlx_amd.train(
        samples=["an apple a day", "keeps the doctor away", "testing one two", "three four"],
        eval_prompts=["an apple", "keeps"] * 2,
        config=cfg,
    )
    assert trainer.iter_count == 2


def test_sft_dialog_pairs(tmp_path):
    cfg = _tiny_model_cfg(default_sft_config(), tmp_path)
    cfg.train.save_best = False
    trainer = trlx_amd.train(
        samples=[["question?", "answer!"], ["other question?", "other answer!"]],
        eval_prompts=["question?"] * 4,
        config=cfg,
    )
    assert trainer.iter_count == 2


def test_rft_end_to_end(tmp_path):
    from trlx_amd.trainer.rft_trainer import RFTConfig

    cfg = _tiny_model_cfg(default_sft_config(), tmp_path)
    cfg.train.trainer = "RFTTrainer"
    cfg.train.save_best = False
    cfg.method = RFTConfig(
        name="RFTConfig",
        gen_kwargs=dict(max_new_tokens=4, top_k=0, top_p=1.0, do_sample=True),
        n_generations_per_prompt=2,
        start_percentile=0.5,
        end_percentile=0.9,
        n_improve_steps=2,
    )

    def reward_fn(samples, prompts, outputs, **kw):
        return [float(len(o)) for o in outputs]

    trainer = trlx_amd.train(
        reward_fn=reward_fn,
        prompts=["aa", "bb", "cc", "dd"],
        eval_prompts=["aa", "bb"] * 2,
        config=cfg,
    )
    assert trainer.iter_count == 2


def test_ppo_resume_from_checkpoint(tmp_path):
    cfg = _tiny_model_cfg(default_ppo_config(), tmp_path)
    cfg.model.num_layers_unfrozen = 1
    cfg.method.num_rollouts = 4
    cfg.method.chunk_size = 4
    cfg.method.ppo_epochs = 1
    cfg.train.save_best = False

    def reward_fn(samples, prompts, outputs, **kw):
        return [1.0 for _ in outputs]

    trlx_amd.train(reward_fn=reward_fn, prompts=["a", "b", "c", "d"],
                   eval_prompts=["a"] * 4, config=cfg)
    ckpt = os.path.join(cfg.train.checkpoint_dir, "checkpoint_2")
    assert os.path.exists(os.path.join(ckpt, "state.pt"))

    cfg2 = _tiny_model_cfg(default_ppo_config(), tmp_path)
    cfg2.model.num_layers_unfrozen = 1
    cfg2.method.num_rollouts = 4
    cfg2.method.chunk_size = 4
    cfg2.method.ppo_epochs = 1
    cfg2.train.save_best = False
    cfg2.train.resume_from_checkpoint = ckpt
    trainer = trlx_amd.train(reward_fn=reward_fn, prompts=["a", "b", "c", "d"],
                             eval_prompts=["a"] * 4, config=cfg2)
    assert trainer.iter_count >= 2


def test_gradient_accumulation_microbatches(tmp_path):
    """minibatch_size < batch_size drives the microbatch accumulation path
    (reference tests/test_trainers.py:161-192)."""
    cfg = _tiny_model_cfg(default_sft_config(), tmp_path, minibatch_size=2)
    cfg.train.save_best = False
    trainer = trlx_amd.train(
        samples=["one", "two", "three", "four", "five", "six", "seven", "eight"],
        eval_prompts=["one"] * 4,
        config=cfg,
    )
    assert trainer.num_mb == 2
    assert trainer.iter_count == 2


def test_ppo_loss_slice_matches_full_width(tmp_path):
    """The logits-slice optimization must reproduce the reference's
    full-width-then-slice loss exactly."""
    from trlx_amd.data.ppo_types import PPORLElement
    from trlx_amd.pipeline.ppo_pipeline import ppo_collate_fn
    from trlx_amd.utils.loading import get_trainer
    from trlx_amd.utils.modeling import logprobs_of_labels

    cfg = _tiny_model_cfg(default_ppo_config(), tmp_path)
    cfg.model.num_layers_unfrozen = 1
    trainer = get_trainer("PPOTrainer")(config=cfg, reward_fn=lambda **kw: [0.0])

    torch.manual_seed(0)
    elems = [
        PPORLElement(
            query_tensor=torch.randint(3, 300, (5,)),
            response_tensor=torch.randint(3, 300, (4,)),
            logprobs=torch.randn(4) - 3,
            values=torch.randn(4) * 0.1,
            rewards=torch.randn(4) * 0.01,
        )
        for _ in range(3)
    ]
    batch = ppo_collate_fn("left", trainer.tokenizer.pad_token_id, elems)
    from trlx_amd.data.ppo_types import PPORLBatch

    batch = PPORLBatch(*[t.to(trainer.device) for t in
                         (batch.query_tensors, batch.response_tensors, batch.logprobs,
                          batch.values, batch.rewards)])
    loss, stats = trainer.loss(batch)

    # independent full-width computation (reference accelerate_ppo_trainer.py:176-192)
    q, r = batch.query_tensors, batch.response_tensors
    response_length = batch.rewards.shape[1]
    advantages, returns = trainer.config.method.get_advantages_and_returns(
        batch.values, batch.rewards, response_length
    )
    tokens = torch.cat((q, r), dim=1)
    attention_mask = tokens.not_equal(trainer.tokenizer.pad_token_id).long()
    with torch.no_grad():
        out = trainer.model(tokens, attention_mask)
    full_logprobs = logprobs_of_labels(out.logits[:, :-1, :], tokens[:, 1:])
    full_values = out.values[:, :-1]
    start = q.shape[1] - 1
    end = start + response_length
    want_loss, _ = trainer.config.method.loss(
        logprobs=full_logprobs[:, start:end],
        values=full_values[:, start:end],
        old_logprobs=batch.logprobs,
        old_values=batch.values,
        advantages=advantages,
        returns=returns,
        mask=attention_mask[:, start + 1 : end + 1],
    )
    # bf16 on GPU: the sliced and full-width lm_head GEMMs reduce in
    # different orders
    atol = 1e-5 if trainer.device.type == "cpu" else 5e-2
    assert torch.allclose(loss.detach(), want_loss.detach(), atol=atol), (loss, want_loss)


def test_gen_kwarg_sweep_eval(tmp_path):
    """A list-valued gen kwarg becomes an eval sweep
    (reference accelerate_base_trainer.py:139-146, 339-360)."""
    cfg = _tiny_model_cfg(default_ilql_config(), tmp_path)
    cfg.train.save_best = False
    cfg.method.gen_kwargs = dict(max_new_tokens=4, top_k=5, beta=[0, 1], temperature=1.0)

    trainer = trlx_amd.train(
        samples=["ab", "cd", "ef", "gh"],
        rewards=[1.0, -1.0, 0.5, -0.5],
        eval_prompts=["ab", "cd"] * 2,
        metric_fn=lambda samples, **kw: {"len": [float(len(s)) for s in samples]},
        config=cfg,
    )
    assert trainer.generate_sweep_kwarg is not None
    assert trainer.iter_count == 2


@pytest.mark.gpu
def test_train_graph_matches_eager(tmp_path, monkeypatch):
    """hipGraph-captured train step matches eager: identical seeds/rollouts
    must yield (near-)identical final weights after several optimizer steps
    (the graph replays the same kernels in the same order)."""
    weights = {}
    for mode in ("graph", "eager"):
        if mode == "eager":
            monkeypatch.setenv("TRLX_AMD_NO_TRAIN_GRAPH", "1")
        else:
            monkeypatch.delenv("TRLX_AMD_NO_TRAIN_GRAPH", raising=False)
        cfg = _tiny_model_cfg(default_ppo_config(), tmp_path / mode)
        # decode kernel needs head_dim >= 32
        cfg.model.model_extra_configs["config"].update(
            num_heads=2, num_kv_heads=2, head_dim=32)
        cfg.train.total_steps = 4
        cfg.train.eval_interval = 100
        cfg.train.checkpoint_interval = 100
        cfg.train.save_best = False
        cfg.model.num_layers_unfrozen = 1
        cfg.method.num_rollouts = 8
        cfg.method.chunk_size = 8
        cfg.method.ppo_epochs = 2
        cfg.train.seed = 1234
        cfg.method.gen_kwargs["seed"] = 7

        def reward_fn(samples, prompts, outputs, **kw):
            return [float(len(o)) for o in outputs]

        trainer = trlx_amd.train(
            reward_fn=reward_fn,
            prompts=["hello", "world", "foo bar", "baz"] * 2,
            eval_prompts=["hello", "sky"],
            config=cfg,
        )
        if mode == "graph":
            assert getattr(trainer, "_train_graphs", None), "graph path was not used"
        weights[mode] = {k: v.detach().float().cpu()
                         for k, v in trainer.model.state_dict().items()}
        del trainer
        torch.cuda.empty_cache()
    for k in weights["graph"]:
        assert torch.allclose(weights["graph"][k], weights["eager"][k], atol=1e-3), k


def test_ppo_update_direction_raises_best_advantage_logprob(tmp_path):
    """Learning-mechanics regression: repeated PPO epochs over a FIXED rollout
    store must raise the policy logprob of the highest-reward element (the
    clipped surrogate's ascent direction).  Guards the sign/masking/slicing of
    the whole logprob -> GAE -> whitened-advantage -> ratio-clip chain."""
    import bench
    from trlx_amd.pipeline import MiniBatchIterator
    from trlx_amd.utils.modeling import logprobs_of_labels

    torch.manual_seed(0)
    bargs = bench.parse_args([])
    bargs.tiny_smoke = True
    bargs.num_rollouts = 16
    bargs.chunk_size = 16
    bargs.batch_size = 8
    bargs.max_new_tokens = 8
    bargs.prompt_len = 4
    bargs.num_prompts = 32
    target = {f"t{i}" for i in range(50, 150)}

    def reward_fn(samples, prompts, outputs, **kwargs):
        return [10.0 * sum(t in target for t in o.split()) / max(len(o.split()), 1)
                for o in outputs]

    trainer, config = bench.build_trainer(bargs, reward_fn=reward_fn)
    for g in trainer.opt.param_groups:
        g["lr"] = 1e-3
    trainer.store.clear_history()
    trainer.make_experience(config.method.num_rollouts)
    batch, lengths = trainer.store.batches[0]
    rsums = torch.stack([batch.rewards[i, : int(lengths[i])].sum()
                         for i in range(len(lengths))])
    bi = int(rsums.argmax())
    assert rsums[bi] > rsums.median(), "need a spread of rewards for the test"
    w = int(lengths[bi])
    q = batch.query_tensors[bi : bi + 1]
    r = batch.response_tensors[bi : bi + 1]

    def best_logprob():
        ids = torch.cat([q, r[:, :w]], 1)
        with torch.no_grad():
            logits = trainer.model(ids, attention_mask=torch.ones_like(ids)).logits
        start = q.shape[1] - 1
        return float(logprobs_of_labels(logits[:, start : start + w],
                                        ids[:, start + 1 : start + 1 + w]).sum())

    before = best_logprob()
    trainer.model.eval()
    for _ in range(6):
        loader = trainer.store.create_loader(8, shuffle=True, seed=1)
        for minibatch in MiniBatchIterator(loader, trainer.mb_size, trainer.num_mb):
            for microbatch in minibatch:
                with trainer._accumulate():
                    loss, _ = trainer.loss(microbatch)
                    loss.backward()
            trainer.reducer.finalize()
            trainer.opt.step()
            trainer.opt.zero_grad()
    after = best_logprob()
    assert after > before + 0.3, (before, after)


def test_ppo_end_to_end_with_ref_offload(tmp_path):
    """model.ref_offload=True (K15): PPO e2e with the frozen reference branch
    offloaded to host memory."""
    cfg = _tiny_model_cfg(default_ppo_config(), tmp_path)
    cfg.model.num_layers_unfrozen = 1
    cfg.model.ref_offload = True
    cfg.method.num_rollouts = 8
    cfg.method.chunk_size = 4
    cfg.method.ppo_epochs = 1

    def reward_fn(samples, prompts, outputs, **kw):
        return [float(len(o)) for o in outputs]

    trainer = trlx_amd.train(
        reward_fn=reward_fn,
        prompts=["hello", "world", "foo", "bar"],
        eval_prompts=["hello"],
        config=cfg,
    )
    assert trainer.model.frozen_head._offloaded
    assert trainer.iter_count == 2
