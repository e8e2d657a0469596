"""Run only the v2 kernel a few times (for PMC counter collection)."""
import os, sys
sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))
import torch
from trlx_amd import ops

ext = ops._require_ext("prof")
N, V, H = 5248, 50257, 768
torch.manual_seed(0)
hidden = (torch.randn(N, H, device="cuda") * 0.5).to(torch.bfloat16)
weight = (torch.randn(V, H, device="cuda") * 0.02).to(torch.bfloat16)
labels = torch.randint(0, V, (N,), device="cuda")
for _ in range(5):
    ext.lm_logprobs_v2(hidden, weight, labels)
torch.cuda.synchronize()
print("done")
