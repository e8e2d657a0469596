"""Standalone timings for the fused decode-stage kernels vs skinny_gemm vs
hipBLASLt on the GPT-2 decode shapes, plus the per-kernel launch floor.

Run on the GPU box: python tools/bench_stage_gemm.py
"""

import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from trlx_amd import ops  # noqa: E402

EXT = ops._load_ext()


def time_fn(fn, reps=200, warmup=20):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = torch.cuda.Event(enable_timing=True)
    t1 = torch.cuda.Event(enable_timing=True)
    t0.record()
    for _ in range(reps):
        fn()
    t1.record()
    torch.cuda.synchronize()
    return t0.elapsed_time(t1) * 1000.0 / reps  # us


def main():
    dev = "cuda"
    shapes = [
        (128, 768, 2304, "qkv"),
        (128, 768, 768, "o"),
        (128, 768, 3072, "fc"),
        (128, 3072, 768, "down"),
        (128, 768, 50257, "lm"),
    ]
    print("per-kernel floor (decode_advance-ish):")
    tok = torch.zeros(128, dtype=torch.long, device=dev)
    outt = torch.zeros(128, 40, dtype=torch.long, device=dev)
    cur = torch.zeros(128, dtype=torch.long, device=dev)
    fin = torch.zeros(128, dtype=torch.bool, device=dev)
    s1 = torch.zeros(1, dtype=torch.long, device=dev)
    s2 = torch.zeros(1, dtype=torch.long, device=dev)
    s3 = torch.zeros(1, dtype=torch.long, device=dev)
    sl = torch.zeros(128, dtype=torch.int32, device=dev)
    pi = torch.zeros(128, dtype=torch.int32, device=dev)
    print(f"  advance: {time_fn(lambda: EXT.decode_advance(tok, outt, cur, fin, s1, s2, s3, sl, pi, None, -1, 0)):8.2f} us")

    for M, K, N, label in shapes:
        a = torch.randn(M, K, device=dev).bfloat16()
        w = (torch.randn(N, K, device=dev) * 0.1).bfloat16()
        bias = torch.zeros(N, device=dev).bfloat16()
        c = torch.empty(M, N, device=dev, dtype=torch.bfloat16)
        nw = torch.ones(K, device=dev).bfloat16()
        nb = torch.zeros(K, device=dev).bfloat16()
        af = a.float()
        nstats = torch.stack([af.sum(-1), (af * af).sum(-1)], -1).contiguous()
        resid = torch.zeros(M, N, device=dev).bfloat16()
        ostats = torch.zeros(M, 2, device=dev)

        t_raw = time_fn(lambda: EXT.stage_gemm(a, w, bias, c, None, None, None, False, 1e-5, 0, None, None))
        t_norm = time_fn(lambda: EXT.stage_gemm(a, w, bias, c, nstats, nw, nb, False, 1e-5, 0, None, None))
        t_full = time_fn(lambda: EXT.stage_gemm(a, w, bias, c, None, None, None, False, 1e-5, 0, resid, ostats))
        t_v2 = time_fn(lambda: EXT.stage_gemm_v2(a, w, bias, c, False, None, None, False, 1e-5, 0, None, None))
        t_v2n = time_fn(lambda: EXT.stage_gemm_v2(a, w, bias, c, True, nw, nb, False, 1e-5, 0, None, None))
        t_v2r = time_fn(lambda: EXT.stage_gemm_v2(a, w, bias, c, False, None, None, False, 1e-5, 0, resid, None))
        ps = torch.zeros(max(K // 16, 1), M, 2, device=dev)
        af2 = a.float()
        ps[0, :, 0] = af2.sum(-1)
        ps[0, :, 1] = (af2 * af2).sum(-1)
        pso = torch.zeros(N // 16 if N % 16 == 0 else 1, M, 2, device=dev)
        t_v3 = time_fn(lambda: EXT.stage_gemm_v3(a, w, bias, c, None, 0, None, None, False, 1e-5, 0, None))
        t_v3n = time_fn(lambda: EXT.stage_gemm_v3(a, w, bias, c, ps, 1, nw, nb, False, 1e-5, 0, None))
        t_v3p = time_fn(lambda: EXT.stage_gemm_v3(a, w, bias, c, None, 0, None, None, False, 1e-5, 0, pso if N % 16 == 0 else None))
        t_sk = time_fn(lambda: EXT.skinny_gemm(a, w, bias, 0))
        t_bl = time_fn(lambda: torch.nn.functional.linear(a, w, bias))
        gbs = (M * K + N * K + M * N) * 2 / 1e9
        wbytes = N * K * 2 / 1e12
        print(f"  {label:5s} M{M} K{K} N{N}: raw {t_raw:7.2f} norm {t_norm:7.2f} "
              f"rs {t_full:7.2f} | v2 {t_v2:7.2f} v2n {t_v2n:7.2f} v2r {t_v2r:7.2f} | "
              f"v3 {t_v3:7.2f} v3n {t_v3n:7.2f} v3p {t_v3p:7.2f} | "
              f"skinny {t_sk:7.2f} blaslt {t_bl:7.2f} us "
              f"(v2 {wbytes / (t_v2 * 1e-6):5.2f} TB/s W-stream)")

    # lm_sample
    M, K, N = 128, 768, 50257
    x = torch.randn(M, K, device=dev).bfloat16()
    w = (torch.randn(N, K, device=dev) * 0.1).bfloat16()
    nw = torch.ones(K, device=dev).bfloat16()
    nb = torch.zeros(K, device=dev).bfloat16()
    xf = x.float()
    nstats = torch.stack([xf.sum(-1), (xf * xf).sum(-1)], -1).contiguous()
    packed = torch.zeros(M, dtype=torch.long, device=dev)
    off = torch.zeros(1, dtype=torch.long, device=dev)

    def lm():
        packed.zero_()
        EXT.lm_sample(x, w, None, nstats, nw, nb, packed, False, 1e-5, 1.0, 7, off)

    print(f"  lm_sample (sampling): {time_fn(lm, reps=50):8.2f} us")

    def lmg():
        packed.zero_()
        EXT.lm_sample(x, w, None, nstats, nw, nb, packed, False, 1e-5, 0.0, 7, off)

    print(f"  lm_sample (greedy):   {time_fn(lmg, reps=50):8.2f} us")

    def lm2():
        packed.zero_()
        EXT.lm_sample_v2(x, w, None, nw, nb, packed, False, 1e-5, 1.0, 7, off)

    print(f"  lm_sample_v2 (sampling): {time_fn(lm2, reps=50):8.2f} us")

    def lm2g():
        packed.zero_()
        EXT.lm_sample_v2(x, w, None, nw, nb, packed, False, 1e-5, 0.0, 7, off)

    print(f"  lm_sample_v2 (greedy):   {time_fn(lm2g, reps=50):8.2f} us")

    xf3 = x.float()
    full = torch.stack([xf3.sum(-1), (xf3 * xf3).sum(-1)], -1)[None].contiguous()

    def lm3():
        packed.zero_()
        EXT.lm_sample_v3(x, w, None, full, 1, nw, nb, packed, False, 1e-5, 1.0, 7, off)

    print(f"  lm_sample_v3 (sampling): {time_fn(lm3, reps=50):8.2f} us")

    def lm3g():
        packed.zero_()
        EXT.lm_sample_v3(x, w, None, full, 1, nw, nb, packed, False, 1e-5, 0.0, 7, off)

    print(f"  lm_sample_v3 (greedy):   {time_fn(lm3g, reps=50):8.2f} us")


if __name__ == "__main__":
    main()
