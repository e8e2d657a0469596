"""Pure-PyTorch reference implementations of every custom op.

These are the numerics ground truth for the HIP kernels (tests compare the
gfx950 kernels against these in fp32) and the CPU execution path for tests on
machines without a GPU.  On a GPU box the dispatcher in ``trlx_amd.ops``
refuses to fall back here unless explicitly allowed — the HIP path must be
the one that runs.
"""

from typing import Optional, Tuple

import torch


def logprobs_of_labels(logits: torch.Tensor, labels: torch.Tensor) -> torch.Tensor:
    """Per-token log-probabilities of ``labels`` under ``logits``.

    Parity: reference trlx/utils/modeling.py:213-219 (``logprobs_of_labels``).
    ``logits``: [..., V]; ``labels``: [...] int64 -> [...] float.
    """
    logprobs = torch.nn.functional.log_softmax(logits.float(), dim=-1)
    return torch.gather(logprobs, -1, labels.unsqueeze(-1)).squeeze(-1)


def rmsnorm(x: torch.Tensor, weight: torch.Tensor, eps: float = 1e-6) -> torch.Tensor:
    """RMSNorm over the last dim (Llama-style)."""
    dtype = x.dtype
    x = x.float()
    var = x.pow(2).mean(-1, keepdim=True)
    x = x * torch.rsqrt(var + eps)
    return (x * weight.float()).to(dtype)


def layernorm(
    x: torch.Tensor, weight: torch.Tensor, bias: Optional[torch.Tensor], eps: float = 1e-5
) -> torch.Tensor:
    """LayerNorm over the last dim."""
    dtype = x.dtype
    out = torch.nn.functional.layer_norm(x.float(), (x.shape[-1],), weight.float(), None if bias is None else bias.float(), eps)
    return out.to(dtype)


def rope_cos_sin(
    seq_len: int,
    dim: int,
    base: float = 10000.0,
    device: Optional[torch.device] = None,
    dtype: torch.dtype = torch.float32,
    position_offset: int = 0,
) -> Tuple[torch.Tensor, torch.Tensor]:
    """Host-precomputed RoPE tables ``[seq_len, dim/2]`` (guide: precompute
    trig on host — on-device sinf/cosf turns memory-bound into VALU-bound)."""
    inv_freq = 1.0 / (base ** (torch.arange(0, dim, 2, device=device, dtype=torch.float32) / dim))
    t = torch.arange(position_offset, position_offset + seq_len, device=device, dtype=torch.float32)
    freqs = torch.outer(t, inv_freq)
    return freqs.cos().to(dtype), freqs.sin().to(dtype)


def apply_rope(
    q: torch.Tensor,
    k: torch.Tensor,
    cos: torch.Tensor,
    sin: torch.Tensor,
    interleaved: bool = False,
) -> Tuple[torch.Tensor, torch.Tensor]:
    """Apply rotary embedding to q, k of shape [B, H, T, D].

    ``cos``/``sin``: [T, D/2].  ``interleaved=True`` rotates (x0,x1),(x2,x3)
    pairs (GPT-J/NeoX style); ``False`` rotates (x_i, x_{i+D/2}) halves
    (Llama style).
    """

    def rot(x):
        xf = x.float()
        if interleaved:
            x1 = xf[..., 0::2]
            x2 = xf[..., 1::2]
        else:
            half = x.shape[-1] // 2
            x1 = xf[..., :half]
            x2 = xf[..., half:]
        c = cos.float().view(1, 1, cos.shape[0], cos.shape[1])
        s = sin.float().view(1, 1, sin.shape[0], sin.shape[1])
        o1 = x1 * c - x2 * s
        o2 = x2 * c + x1 * s
        if interleaved:
            out = torch.stack((o1, o2), dim=-1).flatten(-2)
        else:
            out = torch.cat((o1, o2), dim=-1)
        return out.to(x.dtype)

    return rot(q), rot(k)


def gae_advantages_and_returns(
    values: torch.Tensor,
    rewards: torch.Tensor,
    gamma: float,
    lam: float,
    use_whitening: bool = True,
) -> Tuple[torch.Tensor, torch.Tensor]:
    """Generalized Advantage Estimation over the response dimension.

    Parity: reference trlx/models/modeling_ppo.py:136-173
    (``PPOConfig.get_advantages_and_returns``): reverse scan
    delta_t = r_t + gamma * V_{t+1} - V_t, A_t = delta_t + gamma*lam*A_{t+1};
    returns = A + V; advantages whitened.
    ``values``/``rewards``: [B, T] over response tokens only.
    """
    T = rewards.shape[1]
    lastgaelam = torch.zeros_like(rewards[:, 0])
    advantages_reversed = []
    vals = values.float()
    rews = rewards.float()
    for t in reversed(range(T)):
        nextvalues = vals[:, t + 1] if t < T - 1 else torch.zeros_like(vals[:, 0])
        delta = rews[:, t] + gamma * nextvalues - vals[:, t]
        lastgaelam = delta + gamma * lam * lastgaelam
        advantages_reversed.append(lastgaelam)
    advantages = torch.stack(advantages_reversed[::-1], dim=1)
    returns = advantages + vals
    if use_whitening:
        advantages = whiten(advantages)
    return advantages.detach(), returns


def whiten(xs: torch.Tensor, shift_mean: bool = True, distributed: bool = False, group=None) -> torch.Tensor:
    """Whiten to zero mean / unit variance (optionally over a process group).

    Parity: reference trlx/utils/modeling.py:200-210.
    """
    if distributed:
        from ..utils.modeling import get_global_statistics

        mean, var, _ = get_global_statistics(xs, group=group)
    else:
        var, mean = torch.var_mean(xs)
    whitened = (xs - mean) * torch.rsqrt(var + 1e-8)
    if not shift_mean:
        whitened += mean
    return whitened


def sample_token(
    logits: torch.Tensor,
    temperature: float = 1.0,
    top_k: int = 0,
    top_p: float = 1.0,
    generator: Optional[torch.Generator] = None,
) -> torch.Tensor:
    """One sampling step over ``logits`` [B, V] -> token ids [B].

    Greedy when temperature == 0.  Filtering order matches HF: top-k then
    top-p on the temperature-scaled distribution.
    """
    if temperature == 0.0:
        return logits.argmax(dim=-1)
    logits = logits.float() / temperature
    if top_k and top_k > 0 and top_k < logits.shape[-1]:
        kth = torch.topk(logits, top_k, dim=-1).values[..., -1, None]
        logits = logits.masked_fill(logits < kth, float("-inf"))
    if top_p is not None and 0.0 < top_p < 1.0:
        sorted_logits, sorted_idx = torch.sort(logits, descending=True, dim=-1)
        probs = torch.softmax(sorted_logits, dim=-1)
        cum = probs.cumsum(dim=-1)
        # keep the smallest set whose cumulative prob >= top_p (HF semantics:
        # remove tokens with cumulative probability above the threshold,
        # keeping at least the first)
        remove = cum - probs > top_p
        sorted_logits = sorted_logits.masked_fill(remove, float("-inf"))
        logits = torch.full_like(logits, float("-inf")).scatter(-1, sorted_idx, sorted_logits)
    probs = torch.softmax(logits, dim=-1)
    return torch.multinomial(probs, 1, generator=generator).squeeze(-1)


def causal_softmax(
    scores: torch.Tensor, start_pos: int = 0, key_starts: Optional[torch.Tensor] = None
) -> torch.Tensor:
    """Causal-masked softmax over the last dim of [B, H, Tq, Tk] scores.

    Query position i (global position start_pos + i) may attend to key
    positions <= start_pos + i; keys before ``key_starts[b]`` (left padding)
    are masked out.  Fully-masked rows produce zeros.
    """
    dtype = scores.dtype
    Tq, Tk = scores.shape[-2], scores.shape[-1]
    q_pos = torch.arange(Tq, device=scores.device).unsqueeze(-1) + start_pos
    k_pos = torch.arange(Tk, device=scores.device).unsqueeze(0)
    mask = (k_pos > q_pos).view(1, 1, Tq, Tk)
    if key_starts is not None:
        B = scores.shape[0]
        pad = k_pos.view(1, 1, 1, Tk) < key_starts.view(B, 1, 1, 1)
        mask = mask | pad
    scores = scores.float().masked_fill(mask, float("-inf"))
    probs = torch.softmax(scores, dim=-1)
    # rows with no valid keys (fully masked) -> 0 instead of NaN
    probs = torch.nan_to_num(probs, nan=0.0)
    return probs.to(dtype)


def fused_adamw(
    params: list,
    grads: list,
    exp_avgs: list,
    exp_avg_sqs: list,
    master_params: list,
    step: int,
    lr: float,
    beta1: float,
    beta2: float,
    eps: float,
    weight_decay: float,
) -> None:
    """Multi-tensor AdamW with fp32 master weights; params may be bf16 and are
    re-cast from the updated masters."""
    bc1 = 1 - beta1**step
    bc2 = 1 - beta2**step
    for p, g, m, v, mp in zip(params, grads, exp_avgs, exp_avg_sqs, master_params):
        gf = g.float()
        m.mul_(beta1).add_(gf, alpha=1 - beta1)
        v.mul_(beta2).addcmul_(gf, gf, value=1 - beta2)
        mp.mul_(1 - lr * weight_decay)
        denom = (v / bc2).sqrt_().add_(eps)
        mp.addcdiv_(m, denom, value=-lr / bc1)
        p.copy_(mp.to(p.dtype))


def flash_prefill(q, k, v, key_starts=None, start_pos: int = 0, scale: float = 1.0, tk=None):
    """fp32 reference for the flash prefill kernel (masking semantics match
    causal_softmax: key j visible to query i iff key_start <= j <= start_pos+i)."""
    B, Hq, T, D = q.shape
    Hkv, Sk = k.shape[1], k.shape[2]
    tkv = int(tk) if tk else Sk
    kk = k[:, :, :tkv].float()
    vv = v[:, :, :tkv].float()
    if Hq != Hkv:
        rep = Hq // Hkv
        kk = kk.repeat_interleave(rep, dim=1)
        vv = vv.repeat_interleave(rep, dim=1)
    scores = torch.matmul(q.float() * scale, kk.transpose(-1, -2))
    i = torch.arange(T, device=q.device).view(1, 1, T, 1)
    j = torch.arange(tkv, device=q.device).view(1, 1, 1, tkv)
    mask = j <= (start_pos + i)
    if key_starts is not None:
        mask = mask & (j >= key_starts.view(B, 1, 1, 1).to(q.device))
    scores = scores.masked_fill(~mask, float("-inf"))
    probs = torch.softmax(scores, dim=-1)
    probs = torch.nan_to_num(probs, nan=0.0)
    return torch.matmul(probs, vv).to(q.dtype)
