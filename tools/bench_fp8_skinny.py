"""A/B the fp8 weight-only skinny GEMM vs bf16 skinny vs torch(hipBLASLt)
at decode shapes, graph-replayed (the in-graph cost decode actually pays)."""
import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from trlx_amd import ops

ext = ops._load_ext()


def graph_time(fn, iters=50, reps=20):
    for _ in range(10):
        fn()
    torch.cuda.synchronize()
    g = torch.cuda.CUDAGraph()
    with torch.cuda.graph(g):
        for _ in range(iters):
            fn()
    torch.cuda.synchronize()
    s = torch.cuda.Event(True); e = torch.cuda.Event(True)
    s.record()
    for _ in range(reps):
        g.replay()
    e.record(); torch.cuda.synchronize()
    return s.elapsed_time(e) / (iters * reps) * 1000


for M, K, N, tag in [(128, 768, 2304, "qkv"), (128, 768, 768, "proj"),
                     (128, 768, 3072, "fc1"), (128, 3072, 768, "fc2"),
                     (128, 768, 50257, "lm_head"), (128, 1600, 6400, "xl fc1")]:
    x = (torch.randn(M, K, device="cuda") * 0.5).bfloat16()
    w = (torch.randn(N, K, device="cuda") * 0.1).bfloat16()
    q8, s = ops.quantize_fp8_rows(w)
    t_bf = graph_time(lambda: ext.skinny_gemm(x, w, None, 0))
    t_f8 = graph_time(lambda: ext.skinny_gemm_fp8(x, q8, s, None, 0))
    t_t = graph_time(lambda: torch.nn.functional.linear(x, w))
    wmb = N * K * 2 / 1e6
    print(f"{tag:8s} [{M},{K}]x[{K},{N}] W={wmb:6.1f}MB: torch {t_t:7.2f} us  "
          f"skinny-bf16 {t_bf:7.2f} us  skinny-fp8 {t_f8:7.2f} us "
          f"({wmb/2/1e3/t_f8*1e3:.2f} TB/s eff-fp8)")
