"""Minimal kernel-repetition driver for PMC counter collection on the decode
GEMM shapes (run under rocprofv3 --pmc; analysis in profiles/)."""
import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from trlx_amd import ops  # noqa: E402

EXT = ops._load_ext()
dev = "cuda"
which = os.environ.get("PROF_WHICH", "qkv_raw")
M, K, N = 128, 768, 2304
if which.startswith("lm"):
    N = 50257
a = torch.randn(M, K, device=dev).bfloat16()
w = (torch.randn(N, K, device=dev) * 0.1).bfloat16()
c = torch.empty(M, N, device=dev, dtype=torch.bfloat16)
for _ in range(30):
    if which == "qkv_raw" or which == "lm_raw":
        EXT.stage_gemm(a, w, None, c, None, None, None, False, 1e-5, 0, None, None)
    elif which == "qkv_v2":
        EXT.stage_gemm_v2(a, w, None, c, False, None, None, False, 1e-5, 0, None, None)
    elif which == "blaslt":
        torch.nn.functional.linear(a, w)
torch.cuda.synchronize()
print("done", which)
