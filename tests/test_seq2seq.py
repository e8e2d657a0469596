"""Seq2seq (T5) tests: HF equivalence, wrappers, and e2e PPO/ILQL runs
(parity: reference seq2seq classes in modeling_ppo.py / modeling_ilql.py
and the ppo_sentiments_t5 example path)."""

import pytest
import torch

import trlx_amd
from trlx_amd.data.default_configs import default_ilql_config, default_ppo_config
from trlx_amd.models.modeling_seq2seq import (
    AutoModelForSeq2SeqLMWithHydraValueHead,
    AutoModelForSeq2SeqLMWithILQLHeads,
    AutoModelForSeq2SeqLMWithValueHead,
)
from trlx_amd.models.nn.seq2seq import (
    Seq2SeqConfig,
    Seq2SeqTransformer,
    seq2seq_config_from_hf,
    seq2seq_state_dict_from_hf,
    seq2seq_state_dict_to_hf,
)


def tiny_s2s(**over):
    base = dict(vocab_size=300, d_model=64, d_kv=16, num_heads=4, d_ff=128, num_layers=2,
                num_decoder_layers=2, decoder_start_token_id=2, pad_token_id=2, eos_token_id=1)
    base.update(over)
    return Seq2SeqConfig(**base)


@pytest.mark.parametrize("proj", ["relu", "gated-gelu"])
def test_t5_hf_equivalence(proj):
    torch.manual_seed(0)
    transformers = pytest.importorskip("transformers")
    hf_cfg = transformers.T5Config(vocab_size=99, d_model=64, d_kv=16, num_heads=4, d_ff=128,
                                   num_layers=2, num_decoder_layers=2, dropout_rate=0.0,
                                   feed_forward_proj=proj)
    hf = transformers.T5ForConditionalGeneration(hf_cfg).eval()
    cfg = seq2seq_config_from_hf(hf_cfg.to_dict())
    m = Seq2SeqTransformer(cfg).eval()
    missing, unexpected = m.load_state_dict(seq2seq_state_dict_from_hf(cfg, hf.state_dict()),
                                            strict=False)
    assert not unexpected
    ids = torch.randint(2, 99, (2, 11))
    mask = torch.ones_like(ids)
    mask[0, 8:] = 0
    dec = torch.randint(2, 99, (2, 6))
    dec[:, 0] = 0
    with torch.no_grad():
        want = hf(input_ids=ids, attention_mask=mask, decoder_input_ids=dec).logits
        got = m(ids, mask, decoder_input_ids=dec).logits
    assert torch.allclose(want, got, atol=2e-4), (want - got).abs().max()
    back = seq2seq_state_dict_to_hf(cfg, m.state_dict())
    for k, v in back.items():
        assert torch.allclose(v, hf.state_dict()[k], atol=1e-6), k


def test_seq2seq_incremental_decode_matches_full():
    torch.manual_seed(0)
    m = Seq2SeqTransformer(tiny_s2s()).eval()
    ids = torch.randint(3, 300, (2, 7))
    dec = torch.randint(3, 300, (2, 5))
    with torch.no_grad():
        enc = m.encode(ids)
        full, _, _ = m.decode(dec, enc)
        past = None
        steps = []
        for t in range(dec.shape[1]):
            h, past, _ = m.decode(dec[:, t : t + 1], enc, past=past)
            steps.append(h)
    inc = torch.cat(steps, dim=1)
    assert torch.allclose(inc, full, atol=1e-4), (inc - full).abs().max()


def test_seq2seq_hydra_matches_base_at_init():
    torch.manual_seed(0)
    model = AutoModelForSeq2SeqLMWithHydraValueHead(Seq2SeqTransformer(tiny_s2s()),
                                                    num_layers_unfrozen=1)
    model.eval()
    ids = torch.randint(3, 300, (2, 7))
    dec = torch.randint(3, 300, (2, 5))
    dec[:, 0] = 2
    with torch.no_grad():
        out = model(ids, decoder_input_ids=dec, return_ref_logits=True)
    assert torch.allclose(out.ref_logits, out.logits, atol=1e-4), \
        (out.ref_logits - out.logits).abs().max()


def test_seq2seq_save_load_roundtrip(tmp_path):
    torch.manual_seed(0)
    model = AutoModelForSeq2SeqLMWithValueHead(Seq2SeqTransformer(tiny_s2s()))
    with torch.no_grad():
        for p in model.v_head.parameters():
            p.fill_(0.17)
    model.save_pretrained(str(tmp_path / "s2s"))
    loaded = AutoModelForSeq2SeqLMWithValueHead.from_pretrained(str(tmp_path / "s2s"))
    for p in loaded.v_head.parameters():
        assert torch.all(p == 0.17)
    ids = torch.randint(3, 300, (1, 6))
    dec = torch.randint(3, 300, (1, 4))
    with torch.no_grad():
        a = model(ids, decoder_input_ids=dec)
        b = loaded(ids, decoder_input_ids=dec)
    assert torch.allclose(a.logits, b.logits, atol=1e-5)


def _s2s_cfg(base_cfg, tmp_path):
    cfg = base_cfg
    cfg.model.model_path = "tiny-t5"
    cfg.model.model_arch_type = "seq2seq"
    cfg.model.model_extra_configs = {"config": tiny_s2s().to_dict()}
    cfg.model.num_layers_unfrozen = 1
    cfg.tokenizer.tokenizer_path = "byte"
    cfg.train.seq_length = 32
    cfg.train.batch_size = 4
    cfg.train.total_steps = 2
    cfg.train.eval_interval = 2
    cfg.train.checkpoint_interval = 100
    cfg.train.tracker = None
    cfg.train.save_best = False
    cfg.train.checkpoint_dir = str(tmp_path / "ck")
    return cfg


def test_seq2seq_ppo_end_to_end(tmp_path):
    cfg = _s2s_cfg(default_ppo_config(), tmp_path)
    cfg.method.num_rollouts = 4
    cfg.method.chunk_size = 4
    cfg.method.ppo_epochs = 1
    cfg.method.gen_kwargs = dict(max_new_tokens=4, top_k=0, top_p=1.0, do_sample=True)

    trainer = trlx_amd.train(
        reward_fn=lambda samples, **kw: [float(len(s)) for s in samples],
        prompts=["translate a", "translate bb", "translate c", "translate dd"],
        eval_prompts=["translate a"] * 4,
        config=cfg,
    )
    assert trainer.iter_count == 2


def test_seq2seq_ilql_end_to_end(tmp_path):
    cfg = _s2s_cfg(default_ilql_config(), tmp_path)
    cfg.model.num_layers_unfrozen = -1
    cfg.method.gen_kwargs = dict(max_new_tokens=4, top_k=5, beta=1, temperature=1.0)
    trainer = trlx_amd.train(
        samples=[["question a", "answer a"], ["question b", "answer b"],
                 ["question c", "answer c"], ["question d", "answer d"]],
        rewards=[1.0, -1.0, 0.5, -0.5],
        eval_prompts=["question a"] * 4,
        config=cfg,
    )
    assert trainer.iter_count == 2


@pytest.mark.gpu
def test_seq2seq_gpu_forward_and_generate():
    torch.manual_seed(0)
    m = Seq2SeqTransformer(tiny_s2s()).cuda().to(torch.bfloat16).eval()
    ids = torch.randint(3, 300, (2, 9)).cuda()
    mask = torch.ones_like(ids)
    dec = torch.randint(3, 300, (2, 5)).cuda()
    dec[:, 0] = 2
    with torch.no_grad():
        out = m(ids, mask, decoder_input_ids=dec)
    assert out.logits.isfinite().all()
    gen = m.generate(ids, mask, max_new_tokens=6, do_sample=True)
    assert gen.shape[0] == 2 and gen.shape[1] <= 7


def test_seq2seq_value_branch():
    """Trainable T5 value branch (num_value_layers_unfrozen > 0)."""
    from trlx_amd.models.modeling_seq2seq import (
        AutoModelForSeq2SeqLMWithHydraValueHead,
        AutoModelForSeq2SeqLMWithValueHead,
    )
    from trlx_amd.models.nn.seq2seq import Seq2SeqConfig, Seq2SeqTransformer

    torch.manual_seed(5)
    cfg = Seq2SeqConfig(vocab_size=120, d_model=32, d_ff=64, num_heads=2,
                        num_layers=2, num_decoder_layers=2)
    base = Seq2SeqTransformer(cfg)
    m = AutoModelForSeq2SeqLMWithValueHead(base, num_value_layers_unfrozen=1)
    assert m.v_branch is not None
    ids = torch.randint(3, 100, (2, 5))
    dec = torch.randint(3, 100, (2, 4))
    out = m(ids, torch.ones_like(ids), dec, torch.ones_like(dec))
    assert out.values.shape == (2, 4)
    out.values.sum().backward()
    assert any(p.grad is not None and p.grad.abs().sum() > 0
               for p in m.v_branch.blocks.parameters())

    torch.manual_seed(5)
    hm = AutoModelForSeq2SeqLMWithHydraValueHead(
        Seq2SeqTransformer(cfg), num_layers_unfrozen=1, num_value_layers_unfrozen=1)
    hout = hm(ids, torch.ones_like(ids), dec, torch.ones_like(dec), return_ref_logits=True)
    assert hout.ref_logits is not None and hout.values.shape == (2, 4)
