"""Standalone timings of the decode-step kernels at bench shapes, vs their
in-graph averages (r01 profile) — distinguishes kernel-intrinsic cost from
graph-scheduling floor."""
import os, sys, time
import torch
sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))
from trlx_amd import ops

ext = ops._require_ext("x")
B, H, S, D, HID = 128, 12, 104, 64, 768
torch.manual_seed(0)
dev = "cuda"

def t(fn, iters=200, warmup=20):
    for _ in range(warmup): fn()
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(iters): fn()
    torch.cuda.synchronize(); return (time.perf_counter() - t0) / iters * 1e6

# graph-replay version of the same single kernel
def tg(fn, iters=200):
    s = torch.cuda.Stream(); s.wait_stream(torch.cuda.current_stream())
    with torch.cuda.stream(s):
        fn(); fn()
    torch.cuda.current_stream().wait_stream(s)
    g = torch.cuda.CUDAGraph()
    with torch.cuda.graph(g):
        for _ in range(20): fn()
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(iters // 20): g.replay()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / (20 * (iters // 20)) * 1e6

q = (torch.randn(B, H, 1, D, device=dev) * 0.3).bfloat16()
kc = (torch.randn(B, H, 160, D, device=dev) * 0.3).bfloat16()
vc = (torch.randn(B, H, 160, D, device=dev) * 0.3).bfloat16()
sl = torch.full((B,), S, dtype=torch.int32, device=dev)
ss = torch.zeros(B, dtype=torch.int32, device=dev)
f = lambda: ext.attention_decode(q, kc, vc, sl, 0.125, ss)
print(f"attn_decode    standalone {t(f):6.2f}us  graphed {tg(f):6.2f}us (in-bench-graph ~11.6)")

x = (torch.randn(B, HID, device=dev) * 0.5).bfloat16()
w = (torch.randn(HID, device=dev) * 0.1 + 1).bfloat16()
b = (torch.randn(HID, device=dev) * 0.1).bfloat16()
f = lambda: ext.layernorm_fwd(x, w, b, 1e-5, None)
print(f"layernorm_fwd  standalone {t(f):6.2f}us  graphed {tg(f):6.2f}us (in-bench-graph ~7.1)")

logits = (torch.randn(B, 50257, device=dev) * 2).float()
off = torch.zeros(1, dtype=torch.long, device=dev)
f = lambda: ext.gumbel_sample_dev(logits, 1.0, 123, off)
try:
    f()
    print(f"gumbel_dev     standalone {t(f):6.2f}us  graphed {tg(f):6.2f}us")
except Exception as e:
    print("gumbel probe skipped:", str(e)[:80])

wq = (torch.randn(2304, HID, device=dev) * 0.02).bfloat16()
f = lambda: torch.nn.functional.linear(x, wq)
print(f"qkv blaslt     standalone {t(f):6.2f}us  graphed {tg(f):6.2f}us (in-bench-graph ~9.3)")

# fc_in + gelu: skinny(fused) vs blaslt+gelu, both GRAPHED (in-graph algos differ)
wf = (torch.randn(3072, HID, device=dev) * 0.02).bfloat16()
bf = (torch.randn(3072, device=dev) * 0.1).bfloat16()
f_blaslt = lambda: torch.nn.functional.gelu(torch.nn.functional.linear(x, wf, bf), approximate="tanh")
f_skinny = lambda: ext.skinny_gemm(x, wf, bf, 2)
print(f"fc_in+gelu blaslt graphed {tg(f_blaslt):6.2f}us | skinny graphed {tg(f_skinny):6.2f}us")
wq = (torch.randn(2304, HID, device=dev) * 0.02).bfloat16()
bq = (torch.randn(2304, device=dev) * 0.1).bfloat16()
f_bl_q = lambda: torch.nn.functional.linear(x, wq, bq)
f_sk_q = lambda: ext.skinny_gemm(x, wq, bq, 0)
print(f"qkv blaslt graphed {tg(f_bl_q):6.2f}us | skinny graphed {tg(f_sk_q):6.2f}us")
wo = (torch.randn(HID, HID, device=dev) * 0.02).bfloat16()
f_blaslt2 = lambda: torch.nn.functional.linear(x, wo)
f_skinny2 = lambda: ext.skinny_gemm(x, wo, None, 0)
print(f"o_proj blaslt graphed {tg(f_blaslt2):6.2f}us | skinny graphed {tg(f_skinny2):6.2f}us")
