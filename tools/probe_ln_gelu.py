"""Probe: (1) wave-per-row norm fwd timing vs shapes, (2) does
torch._addmm_activation dispatch a single fused GEMM+bias+GELU kernel on
ROCm/hipBLASLt, and is it faster than addmm + eager gelu on decode shapes?"""
import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from trlx_amd import ops

dev = "cuda"


def timeit(fn, iters=200):
    for _ in range(20):
        fn()
    torch.cuda.synchronize()
    s = torch.cuda.Event(True); e = torch.cuda.Event(True)
    s.record()
    for _ in range(iters):
        fn()
    e.record(); torch.cuda.synchronize()
    return s.elapsed_time(e) / iters * 1000  # us


print("== norm fwd timings (wave dispatch for H<=4096) ==")
for N, H in [(128, 768), (128, 1600), (512, 768), (3328, 768), (128, 4096), (128, 6144)]:
    x = torch.randn(N, H, device=dev).bfloat16()
    r = torch.randn(N, H, device=dev).bfloat16()
    w = torch.randn(H, device=dev).bfloat16()
    b = torch.randn(H, device=dev).bfloat16()
    t_ln = timeit(lambda: ops.layernorm_add(x, r, w, b, 1e-5))
    t_rms = timeit(lambda: ops.rmsnorm_add(x, r, w, 1e-6))
    print(f"N={N:5d} H={H:5d}: layernorm {t_ln:7.2f} us  rmsnorm {t_rms:7.2f} us")

print("== gelu fusion probe ==")
has = hasattr(torch, "_addmm_activation")
print("torch._addmm_activation exists:", has)
if has:
    for N, K, M in [(128, 768, 3072), (5248, 768, 3072), (128, 1600, 6400)]:
        x = torch.randn(N, K, device=dev).bfloat16()
        W = torch.randn(M, K, device=dev).bfloat16()
        bias = torch.randn(M, device=dev).bfloat16()
        ref = torch.nn.functional.gelu(torch.addmm(bias, x, W.t()))
        try:
            out = torch._addmm_activation(bias, x, W.t(), use_gelu=True)
            err = (out.float() - ref.float()).abs().max().item()
            t_f = timeit(lambda: torch._addmm_activation(bias, x, W.t(), use_gelu=True), 100)
            t_e = timeit(lambda: torch.nn.functional.gelu(torch.addmm(bias, x, W.t())), 100)
            t_tanh = (torch.nn.functional.gelu(torch.addmm(bias, x, W.t()), approximate="tanh")
                      .float() - out.float()).abs().max().item()
            print(f"[{N},{K}]x[{K},{M}] fused {t_f:8.2f} us  eager {t_e:8.2f} us  "
                  f"maxerr(exact) {err:.2e}  maxerr(tanh) {t_tanh:.2e}")
        except Exception as ex:
            print("  addmm_activation failed:", ex)
