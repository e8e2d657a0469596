"""Ablation sweep for lm_logprobs_v2 (TRLX_AMD_LMLP_MODE bits)."""
import os
import sys
import time

import torch

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))
from trlx_amd import ops  # noqa: E402

ext = ops._require_ext("x")
N, V, H = 5248, 50257, 768
torch.manual_seed(0)
h = (torch.randn(N, H, device="cuda") * 0.5).bfloat16()
w = (torch.randn(V, H, device="cuda") * 0.02).bfloat16()
l = torch.randint(0, V, (N,), device="cuda")
for _ in range(5):
    ext.lm_logprobs_v2(h, w, l)
torch.cuda.synchronize()
t0 = time.perf_counter()
for _ in range(20):
    ext.lm_logprobs_v2(h, w, l)
torch.cuda.synchronize()
us = (time.perf_counter() - t0) / 20 * 1e6
mode = os.environ.get("TRLX_AMD_LMLP_MODE", "0")
print("mode=%s: %8.1f us" % (mode, us))
