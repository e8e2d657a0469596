"""Microbench: skinny_gemm vs F.linear on decode shapes."""
import os, sys, time
import torch
sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))
from trlx_amd import ops

ext = ops._require_ext("x")
for (M, N, K, act) in [(128, 2304, 768, 0), (128, 768, 768, 0), (128, 3072, 768, 2),
                       (128, 768, 3072, 0), (128, 50257, 768, 0)]:
    torch.manual_seed(0)
    x = (torch.randn(M, K, device="cuda") * 0.5).bfloat16()
    w = (torch.randn(N, K, device="cuda") * 0.02).bfloat16()
    b = (torch.randn(N, device="cuda") * 0.1).bfloat16()
    def t(fn, iters=50):
        for _ in range(10): fn()
        torch.cuda.synchronize(); t0 = time.perf_counter()
        for _ in range(iters): fn()
        torch.cuda.synchronize(); return (time.perf_counter() - t0) / iters * 1e6
    t_sk = t(lambda: ext.skinny_gemm(x, w, b, act))
    t_bl = t(lambda: torch.nn.functional.linear(x, w, b))
    if act: t_bl_act = t(lambda: torch.nn.functional.gelu(torch.nn.functional.linear(x, w, b), approximate="tanh"))
    else: t_bl_act = t_bl
    bw = N * K * 2 / t_sk / 1e3  # GB/s of weight stream
    print(f"M{M} N{N} K{K} act{act}: skinny {t_sk:7.2f}us ({bw:6.0f} GB/s W) vs blaslt(+act) {t_bl_act:7.2f}us  speedup {t_bl_act/t_sk:.2f}x")
