"""Property-based tests (hypothesis) for the invariants the reference's
behavior depends on: dialogue tokenization truncation rules, config
round-trips, GAE math, whitening, and the PPO collate shapes."""

import hypothesis.strategies as st
import torch
from hypothesis import given, settings

from trlx_amd import TRLConfig
from trlx_amd.data.default_configs import default_ppo_config
from trlx_amd.ops import reference
from trlx_amd.pipeline.offline_pipeline import tokenize_dialogue
from trlx_amd.utils.tokenizer import ByteTokenizer

SETTINGS = dict(max_examples=25, deadline=None)

words = st.text(alphabet="abcdefgh ", min_size=1, max_size=12)


@settings(**SETTINGS)
@given(st.lists(words, min_size=2, max_size=6).map(lambda l: l[: len(l) // 2 * 2]),
       st.integers(min_value=3, max_value=48),
       st.sampled_from(["left", "right"]))
def test_tokenize_dialogue_invariants(dialogue, max_length, trunc_side):
    if not dialogue:
        dialogue = ["a", "b"]
    tok = ByteTokenizer(truncation_side=trunc_side)
    msgs = tokenize_dialogue(dialogue, tok, max_length=max_length)
    total = sum(len(m.tokens) for m in msgs)
    # bound: max_length plus a possible BOS insertion
    assert total <= max_length + 1
    assert not msgs[0].is_output
    if trunc_side == "left":
        # left truncation preserves the tail: the output and its eos survive
        assert msgs[-1].is_output
        assert msgs[-1].tokens[-1] == tok.eos_token_id
    else:
        # right truncation keeps the head; the output may be truncated away
        # entirely (reference offline_pipeline semantics) — what remains must
        # be a prefix of the untruncated stream
        full = tokenize_dialogue(dialogue, tok, max_length=10**6)
        flat = [t for m in msgs for t in m.tokens]
        full_flat = [t for m in full for t in m.tokens]
        assert flat == full_flat[: len(flat)]


@settings(**SETTINGS)
@given(st.integers(min_value=0, max_value=2**31 - 1))
def test_config_update_roundtrip(seed):
    import random

    rng = random.Random(seed)
    cfg = default_ppo_config()
    overrides = {
        "method.ppo_epochs": rng.randint(1, 10),
        "train.batch_size": rng.choice([2, 4, 8, 32]),
        "method.gen_kwargs": dict(max_new_tokens=rng.randint(1, 64)),
        "train.seed": rng.randint(0, 10**6),
    }
    updated = TRLConfig.update(cfg.to_dict(), overrides)
    assert updated.method.ppo_epochs == overrides["method.ppo_epochs"]
    assert updated.train.batch_size == overrides["train.batch_size"]
    assert updated.method.gen_kwargs["max_new_tokens"] == \
        overrides["method.gen_kwargs"]["max_new_tokens"]
    # to_dict/from_dict stability
    d = updated.to_dict()
    assert TRLConfig.from_dict(d).to_dict() == d


@settings(**SETTINGS)
@given(st.integers(min_value=1, max_value=6), st.integers(min_value=1, max_value=24),
       st.floats(min_value=0.0, max_value=1.0), st.floats(min_value=0.0, max_value=1.0),
       st.integers(min_value=0, max_value=2**31 - 1))
def test_gae_matches_naive(B, T, gamma, lam, seed):
    g = torch.Generator().manual_seed(seed)
    values = torch.randn(B, T, generator=g)
    rewards = torch.randn(B, T, generator=g)
    adv, ret = reference.gae_advantages_and_returns(values, rewards, gamma, lam,
                                                    use_whitening=False)
    want = torch.zeros(B, T)
    for b in range(B):
        last = 0.0
        for t in reversed(range(T)):
            nextv = values[b, t + 1] if t < T - 1 else 0.0
            delta = rewards[b, t] + gamma * nextv - values[b, t]
            last = delta + gamma * lam * last
            want[b, t] = last
    assert torch.allclose(adv, want, atol=1e-4)
    assert torch.allclose(ret, want + values, atol=1e-4)


@settings(**SETTINGS)
@given(st.integers(min_value=8, max_value=512), st.integers(min_value=0, max_value=2**31 - 1))
def test_whiten_properties(n, seed):
    g = torch.Generator().manual_seed(seed)
    xs = torch.randn(n, generator=g) * 7 + 3
    w = reference.whiten(xs)
    assert abs(w.mean().item()) < 1e-4
    assert abs(w.float().var(unbiased=True).item() - 1.0) < 0.02
    w2 = reference.whiten(xs, shift_mean=False)
    assert abs((w2 - w).std().item()) < 1e-4  # differ only by a constant shift


@settings(**SETTINGS)
@given(st.lists(st.tuples(st.integers(1, 8), st.integers(1, 8)), min_size=1, max_size=6))
def test_ppo_collate_shapes(lengths):
    from trlx_amd.data.ppo_types import PPORLElement
    from trlx_amd.pipeline.ppo_pipeline import ppo_collate_fn

    elems = [
        PPORLElement(torch.arange(q) + 3, torch.arange(r) + 3, torch.zeros(r),
                     torch.zeros(r), torch.zeros(r))
        for q, r in lengths
    ]
    batch = ppo_collate_fn("left", 0, elems)
    qmax = max(q for q, _ in lengths)
    rmax = max(r for _, r in lengths)
    assert batch.query_tensors.shape == (len(lengths), qmax)
    assert batch.response_tensors.shape == (len(lengths), rmax)
    assert batch.logprobs.shape == batch.values.shape == batch.rewards.shape \
        == (len(lengths), rmax)
    # queries left-padded: the LAST q tokens of each row are the original
    for i, (q, r) in enumerate(lengths):
        assert batch.query_tensors[i, qmax - q:].tolist() == (torch.arange(q) + 3).tolist()


@settings(max_examples=30, deadline=None)
@given(st.integers(2, 30), st.floats(0.1, 0.99), st.integers(0, 2**31 - 1))
def test_top_p_filter_keeps_nucleus(v, top_p, seed):
    """CPU reference sampling under top-p only ever emits tokens from the
    nucleus: the smallest prefix of the sorted distribution whose exclusive
    cumulative mass is <= top_p (HF semantics)."""
    import torch

    from trlx_amd.ops import reference

    g = torch.Generator().manual_seed(seed)
    logits = torch.randn(1, v, generator=g) * 3
    probs = torch.softmax(logits[0], -1)
    sp, si = torch.sort(probs, descending=True)
    cum = sp.cumsum(0)
    keep = {int(si[i]) for i in range(v) if float(cum[i] - sp[i]) <= top_p}
    for off in range(20):
        tok = int(reference.sample_token(logits, 1.0, 0, top_p,
                                         generator=torch.Generator().manual_seed(off)))
        assert tok in keep, (tok, keep, top_p)


@settings(max_examples=30, deadline=None)
@given(st.integers(1, 6), st.integers(2, 12), st.integers(3, 40), st.integers(1, 8),
       st.integers(0, 2**31 - 1))
def test_batched_index_select_matches_expanded_gather(b, t, h, na, seed):
    """The advanced-indexing fast path equals the reference expanded-index
    gather for any shape, forward and backward."""
    import torch

    from trlx_amd.models.modeling_ilql import batched_index_select

    g = torch.Generator().manual_seed(seed)
    x = torch.randn(b, t, h, generator=g, requires_grad=True)
    x2 = x.detach().clone().requires_grad_(True)
    idx = torch.randint(0, t, (b, na), generator=g)
    fast = batched_index_select(x, idx)
    ref = x2.gather(1, idx.unsqueeze(-1).expand(b, na, h))
    assert torch.equal(fast, ref)
    go = torch.randn(b, na, h, generator=g)
    fast.backward(go)
    ref.backward(go)
    assert torch.equal(x.grad, x2.grad)
