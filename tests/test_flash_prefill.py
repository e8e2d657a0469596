"""Numerics tests for the forward-only flash prefill attention kernel
(csrc/flash_prefill.hip) against the fp32 torch reference, plus the
engine-level check that the no_grad experience forward with the flash path
matches the materializing path."""

import os

import pytest
import torch

pytestmark = pytest.mark.gpu

if torch.cuda.is_available():
    from trlx_amd import ops
    from trlx_amd.ops import reference

    EXT = ops._load_ext()


@pytest.mark.parametrize("D", [64, 128])
@pytest.mark.parametrize("T", [17, 64, 200])
def test_flash_prefill_vs_reference(D, T):
    torch.manual_seed(0)
    B, Hq = 3, 4
    dev = "cuda"
    q = (torch.randn(B, Hq, T, D, device=dev) * 0.5).bfloat16()
    k = (torch.randn(B, Hq, T, D, device=dev) * 0.5).bfloat16()
    v = torch.randn(B, Hq, T, D, device=dev).bfloat16()
    key_starts = torch.tensor([0, 3, 7], device=dev, dtype=torch.int32)
    scale = 1.0 / D ** 0.5

    out = ops.flash_prefill(q, k, v, key_starts, 0, scale)
    ref = reference.flash_prefill(q.float(), k.float(), v.float(), key_starts, 0, scale)
    torch.testing.assert_close(out.float(), ref.float(), atol=3e-2, rtol=3e-2)


def test_flash_prefill_gqa_and_cache_stride():
    torch.manual_seed(1)
    B, Hq, Hkv, T, D, Sk = 2, 8, 2, 40, 64, 96
    dev = "cuda"
    q = (torch.randn(B, Hq, T, D, device=dev) * 0.5).bfloat16()
    kc = torch.zeros(B, Hkv, Sk, D, device=dev).bfloat16()
    vc = torch.zeros(B, Hkv, Sk, D, device=dev).bfloat16()
    kc[:, :, :T] = (torch.randn(B, Hkv, T, D, device=dev) * 0.5).bfloat16()
    vc[:, :, :T] = torch.randn(B, Hkv, T, D, device=dev).bfloat16()
    ks = torch.tensor([0, 5], device=dev, dtype=torch.int32)
    out = ops.flash_prefill(q, kc, vc, ks, 0, 0.125, tk=T)
    ref = reference.flash_prefill(q.float(), kc.float(), vc.float(), ks, 0, 0.125, tk=T)
    torch.testing.assert_close(out.float(), ref.float(), atol=3e-2, rtol=3e-2)


def test_model_forward_flash_matches_materializing():
    """The no_grad model forward must produce the same logits with the flash
    path on and off (TRLX_AMD_NO_FLASH_PREFILL=1 -> [B,H,T,T] path)."""
    from trlx_amd.models.nn.config import TransformerConfig
    from trlx_amd.models.nn.transformer import CausalTransformer

    torch.manual_seed(2)
    cfg = TransformerConfig(vocab_size=500, hidden_size=128, num_layers=3, num_heads=2,
                            max_position_embeddings=256, arch_name="gpt2")
    model = CausalTransformer(cfg).cuda().bfloat16().eval()
    ids = torch.randint(3, 500, (4, 130), device="cuda")
    mask = torch.ones_like(ids)
    mask[0, :9] = 0
    with torch.no_grad():
        flash_logits = model(ids, attention_mask=mask).logits
        os.environ["TRLX_AMD_NO_FLASH_PREFILL"] = "1"
        try:
            mat_logits = model(ids, attention_mask=mask).logits
        finally:
            del os.environ["TRLX_AMD_NO_FLASH_PREFILL"]
    torch.testing.assert_close(flash_logits.float(), mat_logits.float(), atol=8e-2, rtol=8e-2)


def test_generation_prefill_flash_consistent():
    """Greedy generation (flash prefill fills the cache) matches generation
    with the materializing prefill."""
    from trlx_amd.models.nn.config import TransformerConfig
    from trlx_amd.models.nn.generation import generate
    from trlx_amd.models.nn.transformer import CausalTransformer

    torch.manual_seed(3)
    cfg = TransformerConfig(vocab_size=400, hidden_size=128, num_layers=2, num_heads=2,
                            max_position_embeddings=128, arch_name="llama",
                            norm="rmsnorm", position_encoding="rope", activation="silu",
                            swiglu=True, attn_bias=False, mlp_bias=False,
                            intermediate_size=256, tie_word_embeddings=False)
    model = CausalTransformer(cfg).cuda().bfloat16().eval()
    # rope tables must stay fp32 (cast_compute semantics)
    model.rope_cos = model.rope_cos.float()
    model.rope_sin = model.rope_sin.float()
    ids = torch.randint(3, 400, (4, 21), device="cuda")
    mask = torch.ones_like(ids)
    mask[1, :4] = 0
    out_flash = generate(model, ids, mask, max_new_tokens=6, do_sample=False)
    del model._decode_engine
    os.environ["TRLX_AMD_NO_FLASH_PREFILL"] = "1"
    try:
        out_mat = generate(model, ids, mask, max_new_tokens=6, do_sample=False)
    finally:
        del os.environ["TRLX_AMD_NO_FLASH_PREFILL"]
    # single near-tie flips drift the greedy trajectory (see the fused-decode
    # test note); per-step numerics are covered by the reference tests
    agree = (out_flash == out_mat).float().mean().item()
    assert agree >= 0.85, (agree, out_flash, out_mat)
