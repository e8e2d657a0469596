import os, sys, torch
sys.path.insert(0, "/root/repo")
from trlx_amd.models.nn.config import TransformerConfig
from trlx_amd.models.nn.transformer import CausalTransformer

torch.manual_seed(3)
cfg = TransformerConfig(vocab_size=400, hidden_size=128, num_layers=2, num_heads=2,
                        max_position_embeddings=128, arch_name="llama",
                        norm="rmsnorm", position_encoding="rope", activation="silu",
                        swiglu=True, attn_bias=False, mlp_bias=False,
                        intermediate_size=256, tie_word_embeddings=False)
model = CausalTransformer(cfg).cuda().bfloat16().eval()
model.rope_cos = model.rope_cos.float(); model.rope_sin = model.rope_sin.float()
ids = torch.randint(3, 400, (4, 21), device="cuda")
mask = torch.ones_like(ids); mask[1, :4] = 0

def prefill_logits():
    kv = model.new_kv_cache(4, 30, device="cuda")
    with torch.no_grad():
        out = model(ids, attention_mask=mask, kv_cache=kv, start_pos=0, return_logits=False)
        return model.lm_head(out.last_hidden_state[:, -1:, :])[:, 0].float()

lf = prefill_logits()
os.environ["TRLX_AMD_NO_FLASH_PREFILL"] = "1"
lm = prefill_logits()
del os.environ["TRLX_AMD_NO_FLASH_PREFILL"]
d = (lf - lm).abs()
print("cache-prefill logit maxdiff", d.max().item(), "mean", d.mean().item())
print("argmax agree", (lf.argmax(-1) == lm.argmax(-1)).tolist())

# no-cache forward for the same model
with torch.no_grad():
    a = model(ids, attention_mask=mask).logits.float()
    os.environ["TRLX_AMD_NO_FLASH_PREFILL"] = "1"
    b = model(ids, attention_mask=mask).logits.float()
    del os.environ["TRLX_AMD_NO_FLASH_PREFILL"]
print("no-cache logit maxdiff", (a-b).abs().max().item())
