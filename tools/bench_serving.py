"""Serving throughput: continuous batching (slot pool, graph step) vs
one-at-a-time generation and vs a single fixed batch, on flagship GPT-2."""
import sys, os, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from trlx_amd.models.nn.config import preset
from trlx_amd.models.nn.generation import GenerateConfig, generate
from trlx_amd.models.nn.transformer import CausalTransformer
from trlx_amd.serving import ContinuousBatcher

torch.manual_seed(0)
import sys as _sys
_model_name = _sys.argv[1] if len(_sys.argv) > 1 else "gpt2"
m = CausalTransformer(preset(_model_name)).cuda().bfloat16().eval()
print("model:", _model_name)
N, MAXNEW = 32, 40
prompts = [torch.randint(3, 50257, (int(t),)) for t in torch.randint(16, 64, (N,))]
gen = GenerateConfig(do_sample=True, temperature=1.0, eos_token_id=None, seed=7)

# one-at-a-time (what a naive server does per request)
t0 = time.time()
ntok = 0
for p in prompts[:8]:
    out = generate(m, p.unsqueeze(0).cuda(), gen=GenerateConfig(
        max_new_tokens=MAXNEW, do_sample=True, temperature=1.0, eos_token_id=None, seed=7))
    ntok += out.shape[1] - p.numel()
torch.cuda.synchronize()
seq_tps = ntok / (time.time() - t0)
print(f"one-at-a-time: {seq_tps:8.1f} tok/s (8 requests)")

for slots in (4, 8, 16):
    cb = ContinuousBatcher(m, slots=slots, cache_len=128, gen=gen)
    t0 = time.time()
    futs = [cb.submit(p, max_new_tokens=MAXNEW) for p in prompts]
    cb.run_until_idle()
    ntok = sum(len(f.result()) for f in futs)
    dt = time.time() - t0
    print(f"continuous x{slots:2d} slots: {ntok/dt:8.1f} tok/s ({N} requests, "
          f"graph={'yes' if cb.graph is not None else 'no'})")
