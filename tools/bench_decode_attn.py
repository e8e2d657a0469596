"""Isolate fused_decode_attention cost at bench shapes (graph-replayed, so
launch overhead is excluded — the number the in-graph decode step pays)."""
import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from trlx_amd import ops

ext = ops._load_ext()
dev = "cuda"

for B, H, S, D, length in [(128, 12, 144, 64, 104), (128, 25, 144, 64, 104),
                           (128, 12, 144, 64, 32), (32, 12, 144, 64, 104)]:
    qkv = torch.randn(B, 3 * H * D, device=dev).bfloat16()
    kc = torch.randn(B, H, S, D, device=dev).bfloat16()
    vc = torch.randn(B, H, S, D, device=dev).bfloat16()
    seq_lens = torch.full((B,), length, dtype=torch.int32, device=dev)
    cache_idx = torch.tensor([length - 1], dtype=torch.long, device=dev)
    rot = D
    cs = torch.randn(2048, rot // 2, device=dev).abs()
    sn = torch.randn(2048, rot // 2, device=dev).abs()

    def call():
        ext.fused_decode_attention(qkv, kc, vc, seq_lens, None, cs, sn, cache_idx,
                                   rot, False, 0.125)

    for _ in range(10):
        call()
    torch.cuda.synchronize()
    g = torch.cuda.CUDAGraph()
    with torch.cuda.graph(g):
        for _ in range(50):
            call()
    torch.cuda.synchronize()
    s = torch.cuda.Event(True); e = torch.cuda.Event(True)
    s.record()
    for _ in range(20):
        g.replay()
    e.record(); torch.cuda.synchronize()
    us = s.elapsed_time(e) / (50 * 20) * 1000
    kv_mb = 2 * B * H * length * D * 2 / 1e6
    print(f"B={B} H={H} len={length}: {us:6.2f} us  (KV {kv_mb:.0f} MB -> {kv_mb/1e3/us*1e3:.2f} TB/s)")
