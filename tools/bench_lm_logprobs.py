"""Microbench + numerics check for the fused lm_head+logprobs kernels.

Compares (on the PPO experience shapes):
  - unfused: hipBLASLt GEMM -> logits -> fused logprob-gather kernel
  - v1: single-buffered 128x128 MFMA tile kernel
  - v2: 8-phase pipelined 256x256 MFMA kernel

Run on a GPU box:  python tools/bench_lm_logprobs.py
"""

import os
import sys
import time

import torch

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))

from trlx_amd import ops
from trlx_amd.ops import reference


def timeit(fn, iters=20, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6  # us


def main():
    assert torch.cuda.is_available()
    ext = ops._require_ext("bench")
    V, H = 50257, 768
    torch.manual_seed(0)
    for N in (1312, 2624, 5248):
        hidden = (torch.randn(N, H, device="cuda") * 0.5).to(torch.bfloat16)
        weight = (torch.randn(V, H, device="cuda") * 0.02).to(torch.bfloat16)
        labels = torch.randint(0, V, (N,), device="cuda")

        # ground truth (fp32 torch)
        logits = hidden.float() @ weight.float().t()
        want = reference.logprobs_of_labels(logits, labels)

        got_v1 = ext.lm_logprobs(hidden, weight, labels)
        err1 = (got_v1 - want).abs().max().item()
        got_v2 = ext.lm_logprobs_v2(hidden, weight, labels)
        err2 = (got_v2 - want).abs().max().item()

        def unfused():
            lg = torch.nn.functional.linear(hidden, weight)
            return ops.logprobs_of_labels(lg.unsqueeze(0), labels.unsqueeze(0))

        t_unfused = timeit(unfused)
        t_gemm = timeit(lambda: torch.nn.functional.linear(hidden, weight))
        t_v1 = timeit(lambda: ext.lm_logprobs(hidden, weight, labels))
        t_v2 = timeit(lambda: ext.lm_logprobs_v2(hidden, weight, labels))
        fl = 2.0 * N * V * H
        print(f"N={N}: unfused {t_unfused:8.1f}us (gemm {t_gemm:8.1f}us, "
              f"{fl/t_gemm/1e6:6.1f} TF) | v1 {t_v1:8.1f}us ({fl/t_v1/1e6:6.1f} TF, err {err1:.2e}) "
              f"| v2 {t_v2:8.1f}us ({fl/t_v2/1e6:6.1f} TF, err {err2:.2e})")
        flags = []
        if err2 > 2e-1:
            flags.append("V2 NUMERICS FAIL")
        if flags:
            print("  !!! " + ", ".join(flags))


if __name__ == "__main__":
    main()
