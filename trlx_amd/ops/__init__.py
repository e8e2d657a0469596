"""Custom-op dispatch: HIP/CDNA4 kernels on GPU, torch reference on CPU.

Policy (enforced): on a GPU (ROCm) device the hand-written gfx950 kernels are
the ONLY path — if the compiled extension is missing the op raises instead of
silently falling back to eager PyTorch.  On CPU the pure-torch reference
implementations run so the full framework is testable without hardware.

Set ``TRLX_AMD_ALLOW_EAGER=1`` to permit the eager fallback on GPU (debugging
only; never in benchmarks).
"""

import os
from typing import Optional, Tuple

import torch

from . import reference

_EXT = None
_EXT_TRIED = False
_EXT_ERR: Optional[str] = None


def _load_ext():
    """Import the in-tree compiled extension (trlx_amd._C)."""
    global _EXT, _EXT_TRIED, _EXT_ERR
    if _EXT_TRIED:
        return _EXT
    _EXT_TRIED = True
    try:
        from trlx_amd import _C  # built by setup.py build_ext --inplace

        _EXT = _C
    except ImportError as e:  # pragma: no cover - exercised on GPU boxes
        _EXT = None
        _EXT_ERR = str(e)
    return _EXT


def extension_available() -> bool:
    return _load_ext() is not None


def _require_ext(op_name: str):
    ext = _load_ext()
    if ext is None:
        if os.environ.get("TRLX_AMD_ALLOW_EAGER") == "1":
            return None
        raise RuntimeError(
            f"trlx_amd op '{op_name}' needs the HIP extension (trlx_amd._C) on GPU "
            f"but it is not built (import error: {_EXT_ERR}). "
            f"Run `python setup.py build_ext --inplace` (or __graft_entry__.build()). "
            f"Set TRLX_AMD_ALLOW_EAGER=1 to debug with the eager fallback."
        )
    return ext


# --------------------------------------------------------------------------
# logprobs_of_labels
# --------------------------------------------------------------------------


class _LogprobsOfLabels(torch.autograd.Function):
    @staticmethod
    def forward(ctx, logits, labels, ext):
        logprobs, lse = ext.logprobs_fwd(logits, labels)
        ctx.save_for_backward(logits, labels, lse)
        return logprobs

    @staticmethod
    def backward(ctx, grad_out):
        logits, labels, lse = ctx.saved_tensors
        ext = _require_ext("logprobs_of_labels")
        grad_logits = ext.logprobs_bwd(logits, labels, lse, grad_out.contiguous())
        return grad_logits, None, None


def logprobs_of_labels(logits: torch.Tensor, labels: torch.Tensor) -> torch.Tensor:
    """log softmax(logits)[..., labels] without materializing [., V] logprobs.

    Replaces reference trlx/utils/modeling.py:213-219 — the fused HIP kernel
    does the row max/logsumexp reduction and the single-label gather in one
    pass."""
    if logits.is_cuda:
        ext = _require_ext("logprobs_of_labels")
        if ext is not None:
            flat_logits = logits.reshape(-1, logits.shape[-1]).contiguous()
            flat_labels = labels.reshape(-1).contiguous()
            out = _LogprobsOfLabels.apply(flat_logits, flat_labels, ext)
            return out.view(labels.shape)
    return reference.logprobs_of_labels(logits, labels)


# --------------------------------------------------------------------------
# fused LM-head + logprob gather (no [B,T,V] logits materialization)
# --------------------------------------------------------------------------


def lm_logprobs(hidden: torch.Tensor, weight: torch.Tensor, labels: torch.Tensor) -> torch.Tensor:
    """log p(labels) from final hidden states + lm_head weight, computed
    tile-by-tile over the vocab so the [N, V] logits never hit HBM.

    Inference-only (no grad).  hidden [N,H] (bf16), weight [V,H], labels [N].
    """
    if hidden.is_cuda:
        ext = _require_ext("lm_logprobs")
        if ext is not None and hasattr(ext, "lm_logprobs"):
            h = hidden.contiguous()
            if h.shape[1] % 64 == 0 and hasattr(ext, "lm_logprobs_v2") and \
                    os.environ.get("TRLX_AMD_LM_LOGPROBS_V1") != "1":
                # 8-phase pipelined 256x256 MFMA kernel
                return ext.lm_logprobs_v2(h, weight.contiguous(), labels.contiguous())
            return ext.lm_logprobs(h, weight.contiguous(), labels.contiguous())
    logits = hidden.float() @ weight.float().t()
    return reference.logprobs_of_labels(logits, labels)


class _LMLogprobsTrain(torch.autograd.Function):
    """Training-path fused lm_head+logprobs: forward via the pipelined MFMA
    kernel (logits never materialize, saves only the fp32 logsumexp);
    backward recomputes the logits tile-by-tile and emits dlogits directly
    (csrc/ce_backward.hip), then contracts with W / hidden on hipBLASLt for
    dh and dW."""

    @staticmethod
    def forward(ctx, hidden, weight, labels, ext):
        out, lse = ext.lm_logprobs_v2_with_lse(hidden, weight, labels, True)
        ctx.save_for_backward(hidden, weight, labels, lse)
        return out

    @staticmethod
    def backward(ctx, grad_out):
        hidden, weight, labels, lse = ctx.saved_tensors
        ext = _require_ext("lm_logprobs_train")
        dlogits = ext.ce_dlogits(hidden, weight, labels, lse,
                                 grad_out.contiguous().float())
        dh = dlogits @ weight
        dw = dlogits.t() @ hidden if ctx.needs_input_grad[1] else None
        return dh, dw, None, None


def lm_logprobs_train(hidden: torch.Tensor, weight: torch.Tensor,
                      labels: torch.Tensor) -> torch.Tensor:
    """Differentiable log p(labels) from final hidden states + lm_head weight.

    GPU bf16 (H % 64 == 0): fused kernels, the [N, V] logits exist only as
    the bf16 dlogits of the backward.  Otherwise: plain logits path with
    the same math."""
    if (hidden.is_cuda and hidden.dtype == torch.bfloat16
            and weight.dtype == torch.bfloat16 and hidden.shape[-1] % 64 == 0
            and os.environ.get("TRLX_AMD_FUSED_LM_LOGPROBS") != "0"):
        ext = _require_ext("lm_logprobs_train")
        if ext is not None and hasattr(ext, "ce_dlogits"):
            return _LMLogprobsTrain.apply(hidden.contiguous(), weight.contiguous(),
                                          labels.contiguous(), ext)
    logits = torch.nn.functional.linear(hidden, weight)
    return logprobs_of_labels(logits, labels)


ACT_CODES = {"none": 0, "gelu": 1, "gelu_new": 2, "relu": 3, "silu": 4}
_ACT_FNS = {
    0: lambda y: y,
    1: torch.nn.functional.gelu,
    2: lambda y: torch.nn.functional.gelu(y, approximate="tanh"),
    3: torch.nn.functional.relu,
    4: torch.nn.functional.silu,
}


def skinny_linear(x: torch.Tensor, weight: torch.Tensor, bias: Optional[torch.Tensor] = None,
                  act: int = 0) -> torch.Tensor:
    """Inference linear for skinny-M shapes (KV-cached decode): one 16x16
    MFMA tile per wave streaming the weight at line rate, bias + activation
    fused (csrc/skinny_gemm.hip).  Falls back to F.linear (+ activation)
    when autograd is on, on CPU, or for large M.

    OPT-IN (TRLX_AMD_SKINNY=1): wins its standalone microbench 1.2-2.4x,
    but inside the captured decode graph hipBLASLt's algos already run
    these shapes faster (same-box A/B: 749 vs 722 samples/s with the skinny
    path on) — the standalone hipBLASLt baseline was cold-start slow."""
    K = x.shape[-1]
    M = x.numel() // K
    if (x.is_cuda and not torch.is_grad_enabled() and x.dtype == torch.bfloat16
            and weight.dtype == torch.bfloat16 and M <= 256 and K % 32 == 0):
        if os.environ.get("TRLX_AMD_FP8_DECODE") == "1":
            # fp8 (e4m3) WEIGHT-ONLY decode GEMM: halves the weight stream
            # that the decode L2-access wall is proportional to.  Weight-only
            # quantization; activations and accumulation stay bf16/fp32;
            # per-output-row scales applied exactly on the accumulator.
            # OPT-IN: ~0.4% relative logit noise shifts sampled trajectories.
            ext = _require_ext("skinny_gemm_fp8")
            if ext is not None and hasattr(ext, "skinny_gemm_fp8"):
                q = getattr(weight, "_fp8_cache", None)
                if q is None or q[0].shape != weight.shape:
                    weight._fp8_cache = quantize_fp8_rows(weight)
                    q = weight._fp8_cache
                bb = bias.bfloat16() if bias is not None and bias.dtype != torch.bfloat16 \
                    else bias
                y = ext.skinny_gemm_fp8(x.reshape(M, K).contiguous(), q[0], q[1], bb, act)
                return y.view(*x.shape[:-1], weight.shape[0])
        if os.environ.get("TRLX_AMD_SKINNY") == "1":
            ext = _require_ext("skinny_gemm")
            if ext is not None and hasattr(ext, "skinny_gemm"):
                y = ext.skinny_gemm(x.reshape(M, K).contiguous(), weight.contiguous(), bias,
                                    act)
                return y.view(*x.shape[:-1], weight.shape[0])
    if (act in (1, 2, 3) and bias is not None and x.is_cuda
            and not torch.is_grad_enabled() and x.dtype == torch.bfloat16
            and _addmm_act_ok()):
        # single hipBLASLt GEMM with bias+GELU/ReLU epilogue: the eager
        # elementwise GELU was 127 ms of a 3.1 s bench cycle (profile pf3);
        # measured 24.3 -> 19.6 us on the decode MLP shape, 45.6 -> 33.7 us
        # on the experience shape (tools/probe_ln_gelu.py)
        try:
            y = torch._addmm_activation(bias, x.reshape(M, K), weight.t(),
                                        use_gelu=(act != 3))
            return y.view(*x.shape[:-1], weight.shape[0])
        except RuntimeError:
            global _ADDMM_ACT
            _ADDMM_ACT = False
    y = torch.nn.functional.linear(x, weight, bias)
    return _ACT_FNS[act](y)


def quantize_fp8_rows(w: torch.Tensor):
    """Per-output-row e4m3 weight quantization for the fp8 decode GEMM:
    returns (bytes [N, K] uint8, scales [N] fp32) with scale = amax/448."""
    wf = w.detach().float()
    s = (wf.abs().amax(dim=1) / 448.0).clamp(min=1e-12)
    q = (wf / s[:, None]).clamp(-448.0, 448.0).to(torch.float8_e4m3fn)
    return q.view(torch.uint8).contiguous(), s.contiguous()


def dequantize_fp8_rows(q8: torch.Tensor, s: torch.Tensor) -> torch.Tensor:
    """Reference inverse of quantize_fp8_rows (fp32)."""
    return q8.view(torch.float8_e4m3fn).float() * s[:, None]


def refresh_fp8_caches(module: "torch.nn.Module") -> int:
    """Requantize every cached fp8 decode weight IN PLACE (same buffers, so
    captured decode graphs see the fresh values).  Needed between training
    steps and generation: FusedAdamW writes the parameter arenas from a
    kernel, which torch's version counter never sees, so the caches cannot
    self-invalidate.  Returns the number of refreshed weights."""
    n = 0
    for m in module.modules():
        w = getattr(m, "weight", None)
        q = getattr(w, "_fp8_cache", None) if w is not None else None
        if q is not None:
            q8, s = quantize_fp8_rows(w)
            q[0].copy_(q8)
            q[1].copy_(s)
            n += 1
    return n


_ADDMM_ACT: Optional[bool] = None


def _addmm_act_ok() -> bool:
    global _ADDMM_ACT
    if _ADDMM_ACT is None:
        _ADDMM_ACT = (hasattr(torch, "_addmm_activation")
                      and os.environ.get("TRLX_AMD_NO_FUSED_ACT") != "1")
    return _ADDMM_ACT


# --------------------------------------------------------------------------
# norms
# --------------------------------------------------------------------------


class _RMSNorm(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, eps, ext):
        x2d = x.reshape(-1, x.shape[-1]).contiguous()
        out, inv_rms = ext.rmsnorm_fwd(x2d, weight, eps, None)
        ctx.save_for_backward(x2d, weight, inv_rms)
        ctx.x_shape = x.shape
        return out.view(x.shape)

    @staticmethod
    def backward(ctx, grad_out):
        x2d, weight, inv_rms = ctx.saved_tensors
        ext = _require_ext("rmsnorm")
        dx, dw = ext.rmsnorm_bwd(x2d, weight, inv_rms,
                                 grad_out.reshape(x2d.shape).contiguous(), None)
        return dx.view(ctx.x_shape), dw, None, None


def rmsnorm(x: torch.Tensor, weight: torch.Tensor, eps: float = 1e-6) -> torch.Tensor:
    if x.is_cuda:
        ext = _require_ext("rmsnorm")
        if ext is not None:
            return _RMSNorm.apply(x, weight, eps, ext)
    return reference.rmsnorm(x, weight, eps)


class _RMSNormAdd(torch.autograd.Function):
    """Residual-add fused into RMSNorm: (y, s) = (rmsnorm(x + res), x + res).

    The backward adds the residual-stream gradient ``ds`` into dx inside the
    norm-backward kernel, so the whole residual connection costs zero extra
    elementwise kernels.  dx and dres are the SAME tensor (aliasing is safe:
    autograd accumulates grads out-of-place)."""

    @staticmethod
    def forward(ctx, x, res, weight, eps, ext):
        x2d = x.reshape(-1, x.shape[-1]).contiguous()
        r2d = res.reshape(x2d.shape).contiguous()
        out, inv_rms, s = ext.rmsnorm_fwd(x2d, weight, eps, r2d)
        ctx.save_for_backward(s, weight, inv_rms)
        ctx.x_shape = x.shape
        return out.view(x.shape), s.view(x.shape)

    @staticmethod
    def backward(ctx, grad_out, grad_s):
        s2d, weight, inv_rms = ctx.saved_tensors
        ext = _require_ext("rmsnorm")
        gs = grad_s.reshape(s2d.shape).contiguous() if grad_s is not None else None
        dx, dw = ext.rmsnorm_bwd(s2d, weight, inv_rms,
                                 grad_out.reshape(s2d.shape).contiguous(), gs)
        dx = dx.view(ctx.x_shape)
        return dx, dx, dw, None, None


def rmsnorm_add(x: torch.Tensor, res: torch.Tensor, weight: torch.Tensor, eps: float = 1e-6):
    """Returns (rmsnorm(x + res), x + res)."""
    if x.is_cuda:
        ext = _require_ext("rmsnorm")
        if ext is not None:
            return _RMSNormAdd.apply(x, res, weight, eps, ext)
    s = x + res
    return reference.rmsnorm(s, weight, eps), s


class _LayerNorm(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias, eps, ext):
        x2d = x.reshape(-1, x.shape[-1]).contiguous()
        out, mean, invstd = ext.layernorm_fwd(x2d, weight, bias, eps, None)
        ctx.save_for_backward(x2d, weight, mean, invstd)
        ctx.x_shape = x.shape
        ctx.has_bias = bias is not None
        return out.view(x.shape)

    @staticmethod
    def backward(ctx, grad_out):
        x2d, weight, mean, invstd = ctx.saved_tensors
        ext = _require_ext("layernorm")
        dx, dw, db = ext.layernorm_bwd(x2d, weight, mean, invstd,
                                       grad_out.reshape(x2d.shape).contiguous(), None)
        return dx.view(ctx.x_shape), dw, (db if ctx.has_bias else None), None, None


def layernorm(
    x: torch.Tensor, weight: torch.Tensor, bias: Optional[torch.Tensor], eps: float = 1e-5
) -> torch.Tensor:
    if x.is_cuda:
        ext = _require_ext("layernorm")
        if ext is not None:
            return _LayerNorm.apply(x, weight, bias, eps, ext)
    return reference.layernorm(x, weight, bias, eps)


class _LayerNormAdd(torch.autograd.Function):
    """Residual-add fused into LayerNorm (see _RMSNormAdd)."""

    @staticmethod
    def forward(ctx, x, res, weight, bias, eps, ext):
        x2d = x.reshape(-1, x.shape[-1]).contiguous()
        r2d = res.reshape(x2d.shape).contiguous()
        out, mean, invstd, s = ext.layernorm_fwd(x2d, weight, bias, eps, r2d)
        ctx.save_for_backward(s, weight, mean, invstd)
        ctx.x_shape = x.shape
        ctx.has_bias = bias is not None
        return out.view(x.shape), s.view(x.shape)

    @staticmethod
    def backward(ctx, grad_out, grad_s):
        s2d, weight, mean, invstd = ctx.saved_tensors
        ext = _require_ext("layernorm")
        gs = grad_s.reshape(s2d.shape).contiguous() if grad_s is not None else None
        dx, dw, db = ext.layernorm_bwd(s2d, weight, mean, invstd,
                                       grad_out.reshape(s2d.shape).contiguous(), gs)
        dx = dx.view(ctx.x_shape)
        return dx, dx, dw, (db if ctx.has_bias else None), None, None


def layernorm_add(x: torch.Tensor, res: torch.Tensor, weight: torch.Tensor,
                  bias: Optional[torch.Tensor], eps: float = 1e-5):
    """Returns (layernorm(x + res), x + res)."""
    if x.is_cuda:
        ext = _require_ext("layernorm")
        if ext is not None:
            return _LayerNormAdd.apply(x, res, weight, bias, eps, ext)
    s = x + res
    return reference.layernorm(s, weight, bias, eps), s


# --------------------------------------------------------------------------
# RoPE
# --------------------------------------------------------------------------


class _Rope(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, cos, sin, positions, interleaved, rot, ext):
        out = ext.rope_fwd(x.contiguous(), cos, sin, positions, interleaved, False, rot)
        ctx.save_for_backward(cos, sin, positions)
        ctx.interleaved = interleaved
        ctx.rot = rot
        return out

    @staticmethod
    def backward(ctx, grad_out):
        cos, sin, positions = ctx.saved_tensors
        ext = _require_ext("rope")
        dx = ext.rope_fwd(grad_out.contiguous(), cos, sin, positions, ctx.interleaved, True, ctx.rot)
        return dx, None, None, None, None, None, None


def apply_rope(
    q: torch.Tensor,
    k: torch.Tensor,
    cos: torch.Tensor,
    sin: torch.Tensor,
    positions: Optional[torch.Tensor] = None,
    interleaved: bool = False,
    rot: int = 0,
) -> Tuple[torch.Tensor, torch.Tensor]:
    """Rotary embedding on q, k [B, H, T, D].  ``cos/sin``: [P, rot/2] tables
    (host-precomputed — guide Appendix B); ``positions``: [B, T] int32 row
    indices into the tables (required when prompts are left-padded);
    ``rot`` <= D rotates only the first ``rot`` dims (GPT-J/NeoX partial
    rotary), 0 means full D."""
    D = q.shape[-1]
    if rot <= 0:
        rot = D
    if positions is None:
        T = q.shape[2]
        positions = torch.arange(T, device=q.device, dtype=torch.int32).expand(q.shape[0], T).contiguous()
    if q.is_cuda:
        ext = _require_ext("rope")
        if ext is not None:
            positions = positions.to(torch.int32).contiguous()
            qo = _Rope.apply(q, cos, sin, positions, interleaved, rot, ext)
            ko = _Rope.apply(k, cos, sin, positions, interleaved, rot, ext)
            return qo, ko
    # CPU reference: gather per-position tables [B, T, rot/2]
    c = cos[positions.long()]
    s = sin[positions.long()]
    return _rope_ref_positional(q, k, c, s, interleaved, rot)


def _rope_ref_positional(q, k, c, s, interleaved, rot):
    """CPU reference with per-(b,t) tables c,s of [B, T, rot/2]."""

    def rotate(x):
        xf = x.float()
        xr = xf[..., :rot]
        if interleaved:
            x1, x2 = xr[..., 0::2], xr[..., 1::2]
        else:
            half = rot // 2
            x1, x2 = xr[..., :half], xr[..., half:]
        cc = c.float().unsqueeze(1)  # [B, 1, T, rot/2]
        ss = s.float().unsqueeze(1)
        o1 = x1 * cc - x2 * ss
        o2 = x2 * cc + x1 * ss
        if interleaved:
            out = torch.stack((o1, o2), dim=-1).flatten(-2)
        else:
            out = torch.cat((o1, o2), dim=-1)
        if rot < x.shape[-1]:
            out = torch.cat((out, xf[..., rot:]), dim=-1)
        return out.to(x.dtype)

    return rotate(q), rotate(k)


# --------------------------------------------------------------------------
# GAE / whitening
# --------------------------------------------------------------------------


def gae_advantages_and_returns(
    values: torch.Tensor,
    rewards: torch.Tensor,
    gamma: float,
    lam: float,
    use_whitening: bool = True,
    distributed: bool = False,
    group=None,
) -> Tuple[torch.Tensor, torch.Tensor]:
    """GAE reverse scan (+ optional distributed whitening) — SURVEY.md K6."""
    if values.is_cuda:
        ext = _require_ext("gae")
        if ext is not None:
            adv, ret = ext.gae(values.contiguous().float(), rewards.contiguous().float(), gamma, lam)
            if use_whitening:
                adv = whiten(adv, distributed=distributed, group=group)
            return adv.detach(), ret
    with torch.no_grad():
        adv, ret = reference.gae_advantages_and_returns(values, rewards, gamma, lam, use_whitening=False)
        if use_whitening:
            adv = whiten(adv, distributed=distributed, group=group)
        return adv.detach(), ret


def whiten(xs: torch.Tensor, shift_mean: bool = True, distributed: bool = False, group=None) -> torch.Tensor:
    """Zero-mean/unit-var normalization, optionally with global (cross-rank)
    statistics over ``group`` (reference trlx/utils/modeling.py:200-210)."""
    if xs.is_cuda:
        ext = _require_ext("whiten")
        if ext is not None:
            stats = ext.sum_count(xs.contiguous().float())  # [sum, sumsq, count]
            if distributed:
                import torch.distributed as dist

                dist.all_reduce(stats, group=group)
            mean = stats[0] / stats[2]
            var = stats[1] / stats[2] - mean * mean
            return ext.normalize(xs.contiguous().float(), mean, var, shift_mean)
    return reference.whiten(xs, shift_mean, distributed, group)


# --------------------------------------------------------------------------
# sampling
# --------------------------------------------------------------------------


def decode_prep(
    qkv: torch.Tensor,
    k_cache: torch.Tensor,
    v_cache: torch.Tensor,
    cache_idx: torch.Tensor,
    num_heads: int,
    cos: Optional[torch.Tensor] = None,
    sin: Optional[torch.Tensor] = None,
    key_starts: Optional[torch.Tensor] = None,
    rot: int = 0,
    interleaved: bool = False,
) -> torch.Tensor:
    """Fused decode-step QKV prep: split fused qkv [B,1,QKV], RoPE q/k at
    position (*cache_idx - key_starts[b]), append k/v to the cache at
    *cache_idx, return q [B,Hq,1,D].  GPU-only (the eager path covers CPU)."""
    ext = _require_ext("decode_prep")
    return ext.decode_prep(qkv.contiguous(), k_cache, v_cache, cos, sin, key_starts, cache_idx,
                           num_heads, rot, interleaved)


def fused_decode_attention(
    qkv: torch.Tensor,
    k_cache: torch.Tensor,
    v_cache: torch.Tensor,
    seq_lens: torch.Tensor,
    key_starts: Optional[torch.Tensor],
    cos: Optional[torch.Tensor],
    sin: Optional[torch.Tensor],
    cache_idx: torch.Tensor,
    rot: int,
    interleaved: bool,
    scale: float,
) -> torch.Tensor:
    """MHA decode step in ONE kernel: split fused qkv, RoPE q/k, append k/v
    to the cache at *cache_idx, flash-decode attention — out [B,H,1,D].
    GPU-only (the eager path covers CPU); requires Hq == Hkv."""
    ext = _require_ext("fused_decode_attention")
    ks = key_starts.to(torch.int32).contiguous() if key_starts is not None else None
    if cos is not None and cos.dtype != torch.float32:
        cos, sin = cos.float(), sin.float()  # kernels require fp32 trig tables
    return ext.fused_decode_attention(qkv.contiguous(), k_cache, v_cache,
                                      seq_lens.to(torch.int32).contiguous(), ks, cos, sin,
                                      cache_idx, rot, interleaved, scale)


class _QKVPrep(torch.autograd.Function):
    @staticmethod
    def forward(ctx, qkv, num_heads, num_kv_heads, head_dim, cos, sin, pos, qscale, rot,
                interleaved, ext):
        q, k, v = ext.qkv_prep_fwd(qkv, num_heads, num_kv_heads, head_dim, cos, sin, pos,
                                   qscale, rot, interleaved)
        ctx.save_for_backward(*( [cos, sin, pos] if cos is not None else [] ))
        ctx.has_rope = cos is not None
        ctx.qscale = qscale
        ctx.rot = rot
        ctx.interleaved = interleaved
        return q, k, v

    @staticmethod
    def backward(ctx, dq, dk, dv):
        ext = _require_ext("qkv_prep")
        cos = sin = pos = None
        if ctx.has_rope:
            cos, sin, pos = ctx.saved_tensors
        dqkv = ext.qkv_prep_bwd(dq.contiguous(), dk.contiguous(), dv.contiguous(), cos, sin,
                                pos, ctx.qscale, ctx.rot, ctx.interleaved)
        return dqkv, None, None, None, None, None, None, None, None, None, None


def qkv_prep(
    qkv: torch.Tensor,
    num_heads: int,
    num_kv_heads: int,
    head_dim: int,
    cos: Optional[torch.Tensor] = None,
    sin: Optional[torch.Tensor] = None,
    positions: Optional[torch.Tensor] = None,
    qscale: float = 1.0,
    rot: int = 0,
    interleaved: bool = False,
):
    """Fused split + RoPE + q-scale for the T>1 path (GPU-only; autograd-
    aware — backward applies the inverse rotation and re-packs grads)."""
    ext = _require_ext("qkv_prep")
    if positions is not None:
        positions = positions.to(torch.int32).contiguous()
    if cos is not None and cos.dtype != torch.float32:
        cos, sin = cos.float(), sin.float()  # kernels require fp32 trig tables
    return _QKVPrep.apply(qkv.contiguous(), num_heads, num_kv_heads, head_dim, cos, sin,
                          positions, qscale, rot, interleaved, ext)


def sample_token(
    logits: torch.Tensor,
    temperature: float = 1.0,
    top_k: int = 0,
    top_p: float = 1.0,
    generator: Optional[torch.Generator] = None,
    seed: Optional[int] = None,
    offset: int = 0,
) -> torch.Tensor:
    """One fused sampling step [B, V] -> [B].

    On GPU: Gumbel-max categorical sampling in a single pass (no softmax
    materialization), counter-based RNG keyed by (seed, offset, row) for
    per-DP-rank reproducibility.  top_k/top_p pre-filter runs tile-wise in
    the same kernel via threshold select.
    """
    if logits.is_cuda:
        ext = _require_ext("sample_token")
        if ext is not None:
            if temperature == 0.0:
                return torch.argmax(logits, dim=-1)
            if seed is None:
                seed = int(torch.randint(0, 2**31 - 1, (1,), generator=generator).item())
            dev_offset = isinstance(offset, torch.Tensor)
            if top_p is not None and 0.0 < top_p < 1.0:
                # top-p needs a sorted scan: do the filter with library sort,
                # then fused gumbel sampling on the filtered logits.  Every op
                # here is hipGraph-capturable (device-offset RNG included), so
                # top-p decode stays on the captured-graph path.
                # HF order: top-k first, then top-p (matches reference.sample_token)
                def _gumbel(t):
                    if dev_offset:
                        return ext.gumbel_sample_dev(t, 1.0, None, seed, offset)
                    return ext.gumbel_sample(t, 1.0, None, seed, offset)

                lg = logits.float() / max(temperature, 1e-6)
                if top_k and 0 < top_k < logits.shape[-1]:
                    # both filters confined to the [B, top_k] subset: softmax
                    # over the top-k masked vocab equals softmax over the
                    # subset, so the p-filter never touches [B, V]
                    vals, idx = torch.topk(lg, top_k, dim=-1)  # sorted desc
                    probs = torch.softmax(vals, dim=-1)
                    cum = probs.cumsum(dim=-1)
                    vals = vals.masked_fill(cum - probs > top_p, float("-inf"))
                    pos = _gumbel(vals.contiguous())
                    return idx.gather(-1, pos.view(-1, 1)).view(pos.shape)
                # V-wide: per-row threshold from the sorted scan, elementwise
                # mask — no [B, V] scatter (was a 520 us kernel at [128, 50257])
                sorted_logits, _ = torch.sort(lg, descending=True, dim=-1)
                probs = torch.softmax(sorted_logits, dim=-1)
                cum = probs.cumsum(dim=-1)
                keep = (cum - probs <= top_p).sum(-1, keepdim=True).clamp(min=1) - 1
                thr = sorted_logits.gather(-1, keep)
                lg = lg.masked_fill(lg < thr, float("-inf"))
                return _gumbel(lg.contiguous())
            if top_k and 0 < top_k < logits.shape[-1]:
                lg = (logits.float() / max(temperature, 1e-6)).contiguous()
                thr = torch.topk(lg, top_k, dim=-1).values[:, -1].contiguous()
                if dev_offset:
                    return ext.gumbel_sample_dev(lg, 1.0, thr, seed, offset)
                return ext.gumbel_sample(lg, 1.0, thr, seed, offset)
            # fp32 and bf16 logits both sample natively (no cast kernel)
            lg = logits.contiguous() if logits.dtype in (torch.float32, torch.bfloat16) \
                else logits.float().contiguous()
            if dev_offset:
                return ext.gumbel_sample_dev(lg, float(temperature), None, seed, offset)
            return ext.gumbel_sample(lg, float(temperature), None, seed, offset)
    return reference.sample_token(logits, temperature, top_k, top_p, generator)


def _top_p_filter(logits: torch.Tensor, top_p: float) -> torch.Tensor:
    sorted_logits, sorted_idx = torch.sort(logits, descending=True, dim=-1)
    probs = torch.softmax(sorted_logits, dim=-1)
    cum = probs.cumsum(dim=-1)
    remove = cum - probs > top_p
    sorted_logits = sorted_logits.masked_fill(remove, float("-inf"))
    return torch.full_like(logits, float("-inf")).scatter(-1, sorted_idx, sorted_logits)


# --------------------------------------------------------------------------
# attention decode (single new token vs KV cache)
# --------------------------------------------------------------------------


def attention_decode(
    q: torch.Tensor,
    k_cache: torch.Tensor,
    v_cache: torch.Tensor,
    seq_lens: torch.Tensor,
    scale: float,
    seq_starts: Optional[torch.Tensor] = None,
) -> torch.Tensor:
    """Fused decode attention: q [B, Hq, 1, D] vs cache [B, Hkv, S, D]
    with per-row valid key range [seq_starts[b], seq_lens[b]) -> [B, Hq, 1, D]
    (``seq_starts`` handles left-padded prompts).

    Memory-bound flash-decode kernel; online softmax, no
    [B, H, S] score materialization.
    """
    if q.is_cuda:
        ext = _require_ext("attention_decode")
        if ext is not None:
            ss = None if seq_starts is None else seq_starts.to(torch.int32).contiguous()
            return ext.attention_decode(
                q.contiguous(), k_cache.contiguous(), v_cache.contiguous(),
                seq_lens.to(torch.int32).contiguous(), scale, ss,
            )
    # reference: masked SDPA
    B, Hq, _, D = q.shape
    Hkv = k_cache.shape[1]
    S = k_cache.shape[2]
    qf = q.float()
    kf = k_cache.float()
    vf = v_cache.float()
    if Hkv != Hq:
        rep = Hq // Hkv
        kf = kf.repeat_interleave(rep, dim=1)
        vf = vf.repeat_interleave(rep, dim=1)
    scores = torch.matmul(qf, kf.transpose(-1, -2)) * scale  # [B, Hq, 1, S]
    pos = torch.arange(S, device=q.device).view(1, 1, 1, S)
    mask = pos >= seq_lens.view(B, 1, 1, 1)
    if seq_starts is not None:
        mask = mask | (pos < seq_starts.view(B, 1, 1, 1))
    scores = scores.masked_fill(mask, float("-inf"))
    probs = torch.softmax(scores, dim=-1)
    return torch.matmul(probs, vf).to(q.dtype)


# --------------------------------------------------------------------------
# causal softmax (training attention: rocBLAS GEMM + this fused kernel)
# --------------------------------------------------------------------------


class _CausalSoftmax(torch.autograd.Function):
    @staticmethod
    def forward(ctx, scores, start_pos, key_starts, ext):
        probs = ext.causal_softmax_fwd(scores.contiguous(), start_pos, key_starts)
        ctx.save_for_backward(probs)
        return probs

    @staticmethod
    def backward(ctx, grad_out):
        (probs,) = ctx.saved_tensors
        ext = _require_ext("causal_softmax")
        return ext.causal_softmax_bwd(probs, grad_out.contiguous()), None, None, None


def causal_softmax(
    scores: torch.Tensor, start_pos: int = 0, key_starts: Optional[torch.Tensor] = None
) -> torch.Tensor:
    """Causal-masked softmax of attention scores [B, H, Tq, Tk] (fused mask +
    online softmax; SURVEY.md K2's softmax half — GEMMs ride rocBLAS).

    ``key_starts`` [B] marks the first valid key per batch row (left-padding).
    """
    if scores.is_cuda:
        ext = _require_ext("causal_softmax")
        if ext is not None:
            if key_starts is not None:
                key_starts = key_starts.to(torch.int32).contiguous()
            return _CausalSoftmax.apply(scores, start_pos, key_starts, ext)
    return reference.causal_softmax(scores, start_pos, key_starts)


def flash_prefill(q, k, v, key_starts=None, start_pos: int = 0, scale: float = 1.0,
                  tk=None):
    """Forward-only flash attention for prefill / the no_grad experience pass
    (csrc/flash_prefill.hip): q [B,Hq,T,D] (bf16), k/v [B,Hkv,Sk,D] with the
    first ``tk`` tokens valid (cache prefill passes the whole cache), causal +
    left-pad masking via ``key_starts``.  Returns [B,Hq,T,D] bf16; the
    [B,H,T,T] scores never materialize."""
    ext = _require_ext("flash_prefill")
    if ext is None or not q.is_cuda:
        return reference.flash_prefill(q, k, v, key_starts, start_pos, scale, tk)
    ks = key_starts.to(torch.int32).contiguous() if key_starts is not None else None
    return ext.flash_prefill(q.contiguous(), k, v, ks, int(start_pos), float(scale),
                             int(tk) if tk is not None else 0)


class _FlashAttention(torch.autograd.Function):
    """Differentiable flash attention (training prefill, start_pos == 0,
    Tk == T): forward saves the per-row logsumexp, backward recomputes P from
    it tile-by-tile (csrc/flash_backward.hip) — the [B,H,T,T] scores never
    materialize in either direction."""

    @staticmethod
    def forward(ctx, q, k, v, key_starts, scale, ext):
        out, lse = ext.flash_prefill_lse(q, k, v, key_starts, 0, float(scale), 0, True)
        ctx.save_for_backward(q, k, v, out, lse)
        ctx.key_starts = key_starts
        ctx.scale = scale
        ctx.ext = ext
        return out

    @staticmethod
    def backward(ctx, dout):
        q, k, v, out, lse = ctx.saved_tensors
        dq, dk, dv = ctx.ext.flash_prefill_bwd(q, k, v, out, dout, lse, ctx.key_starts,
                                               float(ctx.scale))
        return dq, dk, dv, None, None, None


def flash_attention(q, k, v, key_starts=None, scale: float = 1.0):
    """Differentiable flash attention for the TRAINING forward (full causal
    self-attention, no cache): q [B,Hq,T,D] bf16, k/v [B,Hkv,T,D].  On CPU or
    without the extension, falls back to the fp32 reference (autograd handles
    the backward there)."""
    ext = _require_ext("flash_attention")
    if ext is None or not q.is_cuda:
        return reference.flash_prefill(q, k, v, key_starts, 0, scale).to(q.dtype)
    ks = key_starts.to(torch.int32).contiguous() if key_starts is not None else None
    return _FlashAttention.apply(q.contiguous(), k.contiguous(), v.contiguous(), ks,
                                 float(scale), ext)
