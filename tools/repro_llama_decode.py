"""Isolate the llama-shape (D=128, MHA, RoPE-full) decode kernels."""
import os, sys
import torch
sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))
from trlx_amd import ops
from trlx_amd.ops import reference

ext = ops._require_ext("x")
torch.manual_seed(0)
B, H, S, D = 8, 32, 32, 128
rot = 128
qkv = (torch.randn(B, 1, 3 * H * D, device="cuda") * 0.3).bfloat16()
kc = torch.zeros(B, H, S, D, device="cuda", dtype=torch.bfloat16)
vc = torch.zeros(B, H, S, D, device="cuda", dtype=torch.bfloat16)
cos, sin = reference.rope_cos_sin(S, rot)
cos, sin = cos.cuda(), sin.cuda()
pos = 5
cache_idx = torch.full((1,), pos, dtype=torch.long, device="cuda")
seq_lens = torch.full((B,), pos + 1, dtype=torch.int32, device="cuda")
ks = torch.zeros(B, dtype=torch.int32, device="cuda")
# seed earlier cache rows
kc[:, :, :pos] = (torch.randn(B, H, pos, D, device="cuda") * 0.3).bfloat16()
vc[:, :, :pos] = (torch.randn(B, H, pos, D, device="cuda") * 0.3).bfloat16()
kc2, vc2 = kc.clone(), vc.clone()

print("step 1: decode_prep D=128...", flush=True)
q = ext.decode_prep(qkv.view(B, -1), kc, vc, cos, sin, ks, cache_idx, H, rot, False)
torch.cuda.synchronize(); print("  ok", flush=True)
print("step 2: attention_decode D=128...", flush=True)
o1 = ext.attention_decode(q, kc, vc, seq_lens, 0.0884, ks)
torch.cuda.synchronize(); print("  ok", flush=True)
print("step 3: fused_decode_attention D=128...", flush=True)
o2 = ext.fused_decode_attention(qkv.view(B, -1), kc2, vc2, seq_lens, ks, cos, sin,
                                cache_idx, rot, False, 0.0884)
torch.cuda.synchronize(); print("  ok", flush=True)
err_o = (o1.float() - o2.float()).abs().max().item()
err_k = (kc.float() - kc2.float()).abs().max().item()
err_v = (vc.float() - vc2.float()).abs().max().item()
print(f"fused vs prep+attn: out {err_o:.2e} kcache {err_k:.2e} vcache {err_v:.2e}")
print("step 4: gumbel bf16 V=32000...", flush=True)
lg = (torch.randn(B, 32000, device="cuda") * 2).bfloat16()
off = torch.zeros(1, dtype=torch.long, device="cuda")
t = ext.gumbel_sample_dev(lg, 1.0, None, 123, off)
torch.cuda.synchronize(); print("  ok", t.max().item(), flush=True)
