"""Localize lm_logprobs_v2 numerics failures by sweeping shapes.

H=64  -> single K-tile (no pipelining)     : fails => static layout bug
H=128 -> two K-tiles (first prefetch use)  : fails => pipeline race
V=256 -> single V-tile                     : isolates reduce/edge handling
V=50176 (multiple of 256) vs 50257        : isolates edge-tile masking
two label sets on same inputs              : error moves => label gather
"""

import os
import sys

import torch

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))

from trlx_amd import ops
from trlx_amd.ops import reference


def check(N, V, H, seed=0):
    torch.manual_seed(seed)
    hidden = (torch.randn(N, H, device="cuda") * 0.5).to(torch.bfloat16)
    weight = (torch.randn(V, H, device="cuda") * 0.02).to(torch.bfloat16)
    labels = torch.randint(0, V, (N,), device="cuda")
    labels2 = torch.randint(0, V, (N,), device="cuda")
    ext = ops._require_ext("dbg")
    logits = hidden.float() @ weight.float().t()
    want = reference.logprobs_of_labels(logits, labels)
    want2 = reference.logprobs_of_labels(logits, labels2)
    got = ext.lm_logprobs_v2(hidden, weight, labels)
    got2 = ext.lm_logprobs_v2(hidden, weight, labels2)
    err = (got - want).abs()
    err2 = (got2 - want2).abs()
    bad = (err > 1e-1).nonzero().flatten()
    # lse implied: label_logit - out; label part: compare against true logit
    msg = f"N={N:5d} V={V:5d} H={H:4d}: maxerr {err.max():.3e} bad_rows {len(bad)}/{N}"
    if len(bad):
        b = bad[:8].tolist()
        true_lab_logit = logits[torch.arange(N, device='cuda'), labels]
        lse = torch.logsumexp(logits, -1)
        got_lse_err = ((true_lab_logit - got) - lse).abs()[bad].max()
        msg += f" first_bad {b} lse_err_on_bad {got_lse_err:.3e}"
        msg += f" label_set_dependent={bool(((err>1e-1)!=(err2>1e-1)).any())}"
    print(msg)


def main():
    for (N, V, H) in [
        (256, 256, 64), (256, 256, 128), (256, 256, 768),
        (256, 512, 768), (256, 50176, 768), (256, 50257, 768),
        (1312, 50257, 768), (100, 300, 768), (256, 256, 192),
    ]:
        check(N, V, H)


if __name__ == "__main__":
    main()
