"""Native LoRA tests (parity: reference tests/test_peft.py — training step
runs, adapter save/load, reference-model behavior under peft)."""

import pytest
import torch

import trlx_amd
from trlx_amd.data.default_configs import default_ppo_config, default_sft_config
from trlx_amd.models.lora import LoRALinear, apply_lora, has_lora, lora_disabled, lora_state_dict
from trlx_amd.models.modeling_ppo import AutoModelForCausalLMWithHydraValueHead
from trlx_amd.models.nn.transformer import CausalTransformer

from conftest import tiny_config

PEFT_CFG = {"peft_type": "LORA", "r": 4, "lora_alpha": 8, "target_modules": ["qkv_proj", "o_proj"]}


def test_lora_injection_and_freeze(tiny_cfg):
    m = CausalTransformer(tiny_cfg)
    apply_lora(m, PEFT_CFG)
    assert has_lora(m)
    trainable = [n for n, p in m.named_parameters() if p.requires_grad]
    assert trainable and all("lora_" in n for n in trainable)
    # adapters start as identity (B zero-init)
    ids = torch.randint(3, 300, (1, 6))
    m2 = CausalTransformer(tiny_cfg)
    m2.load_state_dict(
        {k.replace(".base.", "."): v for k, v in m.state_dict().items() if "lora_" not in k},
        strict=False)
    with torch.no_grad():
        assert torch.allclose(m(ids).logits, m2(ids).logits, atol=1e-5)


def test_lora_disabled_context(tiny_cfg):
    torch.manual_seed(0)
    m = CausalTransformer(tiny_cfg)
    apply_lora(m, PEFT_CFG)
    with torch.no_grad():
        for n, p in m.named_parameters():
            if "lora_B" in n:
                p.add_(torch.randn_like(p) * 0.1)
    ids = torch.randint(3, 300, (1, 6))
    with torch.no_grad():
        with_adapter = m(ids).logits
        with lora_disabled(m):
            without = m(ids).logits
        again = m(ids).logits
    assert not torch.allclose(with_adapter, without, atol=1e-4)
    assert torch.allclose(with_adapter, again, atol=1e-6)


def test_lora_ppo_training_and_adapter_roundtrip(tmp_path, tiny_cfg):
    cfg = default_ppo_config()
    cfg.model.model_path = "tiny"
    cfg.model.model_extra_configs = {"config": tiny_cfg.to_dict()}
    cfg.model.num_layers_unfrozen = -1
    cfg.model.peft_config = PEFT_CFG
    cfg.tokenizer.tokenizer_path = "byte"
    cfg.train.seq_length = 32
    cfg.train.batch_size = 4
    cfg.train.total_steps = 2
    cfg.train.eval_interval = 2
    cfg.train.checkpoint_interval = 100
    cfg.train.tracker = None
    cfg.train.save_best = False
    cfg.train.checkpoint_dir = str(tmp_path / "ck")
    cfg.method.num_rollouts = 4
    cfg.method.chunk_size = 4
    cfg.method.ppo_epochs = 1
    cfg.method.gen_kwargs = dict(max_new_tokens=4, top_k=0, top_p=1.0, do_sample=True)

    trainer = trlx_amd.train(
        reward_fn=lambda samples, **kw: [1.0] * len(samples),
        prompts=["aa", "bb", "cc", "dd"],
        eval_prompts=["aa"] * 4,
        config=cfg,
    )
    assert trainer.iter_count == 2
    assert has_lora(trainer.model.base_model)
    assert trainer.ref_model is None  # peft hydra: no separate ref model

    out_dir = str(tmp_path / "saved")
    trainer.model.save_pretrained(out_dir)
    import os

    assert os.path.exists(os.path.join(out_dir, "adapter_model.pt"))
    loaded = AutoModelForCausalLMWithHydraValueHead.from_pretrained(out_dir, peft_config=PEFT_CFG)
    want = lora_state_dict(trainer.model.base_model)
    got = lora_state_dict(loaded.base_model)
    for k in want:
        assert torch.allclose(want[k], got[k], atol=1e-6), k


@pytest.mark.parametrize("ptype", ["PROMPT_TUNING", "PREFIX_TUNING"])
def test_virtual_token_adapters_forward_and_disable(tiny_cfg, ptype):
    """Prompt/prefix tuning: only adapter params train, logits change, and
    adapters_disabled restores the base model exactly (the peft hydra
    reference path)."""
    from trlx_amd.models.lora import adapters_disabled, apply_peft

    torch.manual_seed(0)
    m = CausalTransformer(tiny_cfg)
    ids = torch.randint(3, 300, (2, 7))
    mask = torch.ones_like(ids)
    mask[0, :2] = 0
    with torch.no_grad():
        base_logits = m(ids, attention_mask=mask).logits
    apply_peft(m, {"peft_type": ptype, "num_virtual_tokens": 4})
    trainable = [n for n, p in m.named_parameters() if p.requires_grad]
    assert trainable and all("soft_prompt" in n or "prefix_kv" in n for n in trainable)
    out = m(ids, attention_mask=mask)
    assert out.logits.shape == base_logits.shape
    assert not torch.allclose(out.logits, base_logits, atol=1e-4)
    with adapters_disabled(m), torch.no_grad():
        off = m(ids, attention_mask=mask).logits
    assert torch.allclose(off, base_logits, atol=1e-5)
    out.logits.sum().backward()
    for n, p in m.named_parameters():
        if p.requires_grad:
            assert p.grad is not None, n


@pytest.mark.parametrize("ptype", ["PROMPT_TUNING", "PREFIX_TUNING"])
@pytest.mark.parametrize("method", ["ppo", "sft", "ilql"])
def test_peft_types_training_step(tmp_path, tiny_cfg, ptype, method):
    """{PPO, SFT, ILQL} x {PROMPT_TUNING, PREFIX_TUNING} training runs +
    adapter checkpoint round-trip (reference tests/test_peft.py:36-60)."""
    from trlx_amd.data.default_configs import default_ilql_config

    peft = {"peft_type": ptype, "num_virtual_tokens": 4}
    if method == "ilql":
        cfg = default_ilql_config()
    elif method == "sft":
        cfg = default_sft_config()
    else:
        cfg = default_ppo_config()
    cfg.model.model_path = "tiny"
    cfg.model.model_extra_configs = {"config": tiny_cfg.to_dict()}
    cfg.model.num_layers_unfrozen = -1
    cfg.model.peft_config = peft
    cfg.tokenizer.tokenizer_path = "byte"
    cfg.train.seq_length = 32
    cfg.train.batch_size = 4
    cfg.train.total_steps = 2
    cfg.train.eval_interval = 2
    cfg.train.checkpoint_interval = 2
    cfg.train.tracker = None
    cfg.train.save_best = False
    cfg.train.checkpoint_dir = str(tmp_path / "ck")
    if method == "ppo":
        cfg.method.num_rollouts = 4
        cfg.method.chunk_size = 4
        cfg.method.ppo_epochs = 1
        cfg.method.gen_kwargs = dict(max_new_tokens=4, top_k=0, top_p=1.0, do_sample=True)
        trainer = trlx_amd.train(
            reward_fn=lambda samples, **kw: [float(len(s)) for s in samples],
            prompts=["aa", "bb", "cc", "dd"], eval_prompts=["aa"], config=cfg)
    elif method == "sft":
        cfg.method.gen_kwargs = dict(max_new_tokens=4, do_sample=True)
        trainer = trlx_amd.train(samples=[["q a", "x"], ["q b", "y"], ["q c", "z"],
                                          ["q d", "w"]],
                                 eval_prompts=["q a"], config=cfg)
    else:
        cfg.method.gen_kwargs = dict(max_new_tokens=4, top_k=5, beta=1, temperature=1.0)
        trainer = trlx_amd.train(samples=["ab", "cd", "ef", "gh"], rewards=[0.1, 0.5, 0.9, 0.2],
                                 eval_prompts=["ab"], config=cfg)
    assert trainer.iter_count == 2
    # adapter round-trip through save_pretrained
    import os

    from trlx_amd.models.lora import adapter_state_dict

    out_dir = str(tmp_path / "hf")
    trainer.save_pretrained(out_dir)
    assert os.path.exists(os.path.join(out_dir, "adapter_model.pt"))
    saved = torch.load(os.path.join(out_dir, "adapter_model.pt"), weights_only=True)
    live = adapter_state_dict(trainer.unwrapped_model.base_model)
    assert set(saved) == set(live)
    for k in saved:
        assert torch.allclose(saved[k], live[k])


@pytest.mark.gpu
@pytest.mark.parametrize("ptype", ["LORA", "PROMPT_TUNING", "PREFIX_TUNING"])
def test_gpu_peft_forward_backward_and_generate(tiny_cfg, ptype):
    """PEFT on hardware: adapter forward+backward through the HIP kernel
    paths and generation (virtual tokens force the eager decode loop on
    GPU; LoRA rides the graph engine)."""
    from trlx_amd.models.lora import apply_peft
    from trlx_amd.models.nn.generation import generate

    torch.manual_seed(0)
    m = CausalTransformer(tiny_cfg).cuda().to(torch.bfloat16)
    cfg = ({"peft_type": "LORA", "r": 4, "lora_alpha": 8} if ptype == "LORA"
           else {"peft_type": ptype, "num_virtual_tokens": 4})
    apply_peft(m, cfg)
    ids = torch.randint(3, 300, (2, 9), device="cuda")
    mask = torch.ones_like(ids)
    mask[0, :3] = 0
    out = m(ids, attention_mask=mask)
    out.logits.float().sum().backward()
    grads = [n for n, p in m.named_parameters() if p.requires_grad and p.grad is not None]
    assert grads, "no adapter grads on GPU"
    m.eval()
    gen = generate(m, ids, mask, max_new_tokens=5, do_sample=False)
    assert gen.shape == (2, 14)
