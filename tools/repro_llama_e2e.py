"""Llama-path bisect: tiny llama-config generate+PPO on GPU, then 7B generate."""
import os, sys
import torch
sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))
from trlx_amd.models.nn.config import TransformerConfig, preset
from trlx_amd.models.nn.generation import generate
from trlx_amd.models.nn.transformer import CausalTransformer

torch.manual_seed(0)
tiny = TransformerConfig(vocab_size=1000, hidden_size=256, num_layers=2, num_heads=2,
                         num_kv_heads=2, intermediate_size=512, max_position_embeddings=256,
                         norm="rmsnorm", position_encoding="rope", activation="silu",
                         swiglu=True, attn_bias=False, mlp_bias=False, arch_name="llama")
m = CausalTransformer(tiny).cuda().bfloat16().eval()
m.rope_cos = m.rope_cos.float(); m.rope_sin = m.rope_sin.float()
ids = torch.randint(3, 900, (8, 8), device="cuda")
print("tiny llama generate (graph)...", flush=True)
out = generate(m, ids, max_new_tokens=6, do_sample=True, seed=3)
torch.cuda.synchronize(); print("  ok", out.shape, flush=True)
print("tiny llama fwd+bwd...", flush=True)
m2 = CausalTransformer(tiny).cuda().bfloat16().train()
m2.rope_cos = m2.rope_cos.float(); m2.rope_sin = m2.rope_sin.float()
loss = m2(torch.randint(3, 900, (4, 16), device="cuda")).logits.float().pow(2).mean()
loss.backward()
torch.cuda.synchronize(); print("  ok", flush=True)

print("7B build...", flush=True)
cfg = preset("llama2-7b")
big = CausalTransformer(cfg).cuda().bfloat16().eval()
big.rope_cos = big.rope_cos.float(); big.rope_sin = big.rope_sin.float()
torch.cuda.synchronize(); print("  built", flush=True)
ids = torch.randint(3, 30000, (128, 64), device="cuda")
print("7B generate 8 tokens (graph)...", flush=True)
out = generate(big, ids, max_new_tokens=8, do_sample=True, seed=3)
torch.cuda.synchronize(); print("  ok", out.shape, flush=True)
