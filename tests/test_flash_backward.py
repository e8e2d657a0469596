"""Numerics tests for the flash attention backward (csrc/flash_backward.hip):
dq/dk/dv from the recomputation kernels vs torch autograd through the fp32
reference (same masking semantics), plus the model-level check that a training
step's gradients match the materializing [B,H,T,T] path."""

import os

import pytest
import torch

pytestmark = pytest.mark.gpu

if torch.cuda.is_available():
    from trlx_amd import ops
    from trlx_amd.ops import reference

    EXT = ops._load_ext()


def _ref_grads(q, k, v, key_starts, scale, dout):
    qf = q.float().detach().requires_grad_(True)
    kf = k.float().detach().requires_grad_(True)
    vf = v.float().detach().requires_grad_(True)
    out = reference.flash_prefill(qf, kf, vf, key_starts, 0, scale)
    out.backward(dout.float())
    return out, qf.grad, kf.grad, vf.grad


@pytest.mark.parametrize("D", [64, 128])
@pytest.mark.parametrize("T", [33, 64, 200])
def test_flash_backward_vs_autograd(D, T):
    torch.manual_seed(0)
    B, Hq = 3, 4
    dev = "cuda"
    q = (torch.randn(B, Hq, T, D, device=dev) * 0.5).bfloat16()
    k = (torch.randn(B, Hq, T, D, device=dev) * 0.5).bfloat16()
    v = (torch.randn(B, Hq, T, D, device=dev) * 0.5).bfloat16()
    key_starts = torch.tensor([0, 3, 7], device=dev, dtype=torch.int32)
    scale = 1.0 / D ** 0.5
    dout = (torch.randn(B, Hq, T, D, device=dev) * 0.5).bfloat16()

    qg = q.detach().requires_grad_(True)
    kg = k.detach().requires_grad_(True)
    vg = v.detach().requires_grad_(True)
    out = ops.flash_attention(qg, kg, vg, key_starts, scale)
    out.backward(dout)

    ref_out, rdq, rdk, rdv = _ref_grads(q, k, v, key_starts, scale, dout)
    torch.testing.assert_close(out.float(), ref_out, atol=3e-2, rtol=3e-2)
    torch.testing.assert_close(vg.grad.float(), rdv, atol=8e-2, rtol=8e-2)
    torch.testing.assert_close(kg.grad.float(), rdk, atol=8e-2, rtol=8e-2)
    torch.testing.assert_close(qg.grad.float(), rdq, atol=8e-2, rtol=8e-2)


def test_flash_backward_gqa():
    torch.manual_seed(1)
    B, Hq, Hkv, T, D = 2, 8, 2, 96, 64
    dev = "cuda"
    q = (torch.randn(B, Hq, T, D, device=dev) * 0.5).bfloat16()
    k = (torch.randn(B, Hkv, T, D, device=dev) * 0.5).bfloat16()
    v = (torch.randn(B, Hkv, T, D, device=dev) * 0.5).bfloat16()
    ks = torch.tensor([0, 5], device=dev, dtype=torch.int32)
    dout = (torch.randn(B, Hq, T, D, device=dev) * 0.5).bfloat16()

    qg = q.detach().requires_grad_(True)
    kg = k.detach().requires_grad_(True)
    vg = v.detach().requires_grad_(True)
    out = ops.flash_attention(qg, kg, vg, ks, 0.125)
    out.backward(dout)

    # fp32 autograd reference with repeat_interleave'd GQA (the reference
    # handles the head expansion; its grads fold back over the group)
    qf = q.float().detach().requires_grad_(True)
    kf = k.float().detach().requires_grad_(True)
    vf = v.float().detach().requires_grad_(True)
    ref = reference.flash_prefill(qf, kf, vf, ks, 0, 0.125)
    ref.backward(dout.float())
    torch.testing.assert_close(out.float(), ref, atol=3e-2, rtol=3e-2)
    torch.testing.assert_close(vg.grad.float(), vf.grad, atol=1e-1, rtol=8e-2)
    torch.testing.assert_close(kg.grad.float(), kf.grad, atol=1e-1, rtol=8e-2)
    torch.testing.assert_close(qg.grad.float(), qf.grad, atol=8e-2, rtol=8e-2)


def test_flash_backward_padded_rows_zero_grad():
    """Left-pad queries/keys (pos < key_start) must carry exactly zero grad."""
    torch.manual_seed(2)
    B, H, T, D = 2, 2, 64, 64
    dev = "cuda"
    q = (torch.randn(B, H, T, D, device=dev) * 0.5).bfloat16().requires_grad_(True)
    k = (torch.randn(B, H, T, D, device=dev) * 0.5).bfloat16().requires_grad_(True)
    v = (torch.randn(B, H, T, D, device=dev) * 0.5).bfloat16().requires_grad_(True)
    ks = torch.tensor([11, 0], device=dev, dtype=torch.int32)
    out = ops.flash_attention(q, k, v, ks, 0.125)
    out.backward(torch.ones_like(out))
    assert torch.all(k.grad[0, :, :11] == 0)
    assert torch.all(v.grad[0, :, :11] == 0)
    assert torch.all(q.grad[0, :, :11] == 0)
    assert k.grad[1].abs().sum() > 0


def test_model_training_grads_flash_vs_materializing():
    """One CE training step: parameter grads with the flash training path must
    match the materializing-softmax path."""
    from trlx_amd.models.nn.config import TransformerConfig
    from trlx_amd.models.nn.transformer import CausalTransformer

    torch.manual_seed(3)
    cfg = TransformerConfig(vocab_size=500, hidden_size=128, num_layers=3, num_heads=2,
                            max_position_embeddings=256, arch_name="gpt2")
    model = CausalTransformer(cfg).cuda().bfloat16()
    ids = torch.randint(3, 500, (4, 130), device="cuda")
    mask = torch.ones_like(ids)
    mask[0, :9] = 0

    def step():
        model.zero_grad(set_to_none=True)
        logits = model(ids, attention_mask=mask).logits
        loss = torch.nn.functional.cross_entropy(
            logits[:, :-1].flatten(0, 1).float(), ids[:, 1:].flatten())
        loss.backward()
        return loss.item(), {n: p.grad.detach().float().clone()
                             for n, p in model.named_parameters() if p.grad is not None}

    loss_f, grads_f = step()
    os.environ["TRLX_AMD_NO_FLASH_PREFILL"] = "1"
    try:
        loss_m, grads_m = step()
    finally:
        del os.environ["TRLX_AMD_NO_FLASH_PREFILL"]

    assert abs(loss_f - loss_m) < 5e-2, (loss_f, loss_m)
    assert grads_f.keys() == grads_m.keys() and len(grads_f) > 0
    for n in grads_f:
        gf, gm = grads_f[n], grads_m[n]
        denom = gm.abs().max().clamp_min(1e-4)
        rel = (gf - gm).abs().max() / denom
        assert rel < 0.12, (n, rel.item())
