"""Synthetic TL;DR summarization task (offline stand-in for
CarperAI/openai_summarize_tldr used by the reference
examples/summarize_rlhf/* — no network, so the dataset is generated).

A "post" is a short word sequence whose TOPIC is its first word; the gold
summary after "TL;DR:" is that topic word.  This keeps the three RLHF stages
meaningful and checkable:

- SFT learns to emit the topic after "TL;DR:",
- the reward model learns to prefer summaries containing the topic over
  random words (pairwise comparisons have a recoverable ground truth),
- PPO can be scored both by the trained RM and by the exact oracle.
"""

import random
from typing import List, Tuple

WORDS = [
    "cat", "dog", "sun", "sky", "sea", "red", "big", "old",
    "run", "fly", "ice", "oak", "map", "jam", "fox", "owl",
]

SEP = " TL;DR:"


def make_post(rng: random.Random) -> Tuple[str, str]:
    """Returns (post_with_sep, gold_summary)."""
    topic = rng.choice(WORDS)
    body = [topic] + rng.choices(WORDS, k=rng.randint(3, 6))
    return " ".join(body) + SEP, topic


def make_sft_samples(n: int, seed: int = 0) -> List[List[str]]:
    """[prompt, output] dialogues for the SFT stage."""
    rng = random.Random(seed)
    out = []
    for _ in range(n):
        post, gold = make_post(rng)
        out.append([post, " " + gold])
    return out


def make_comparisons(n: int, seed: int = 1) -> List[Tuple[str, str]]:
    """(chosen, rejected) full texts for pairwise reward-model training.

    Chosen ends with the gold topic word; rejected ends with a different
    random word — same prompt, so the trajectories share a prefix and
    diverge at the summary (exercises the divergence-masked loss).
    """
    rng = random.Random(seed)
    pairs = []
    for _ in range(n):
        post, gold = make_post(rng)
        bad = rng.choice([w for w in WORDS if w != gold])
        pairs.append((post + " " + gold, post + " " + bad))
    return pairs


def make_prompts(n: int, seed: int = 2) -> List[str]:
    rng = random.Random(seed)
    return [make_post(rng)[0] for _ in range(n)]


def oracle_reward(samples: List[str]) -> List[float]:
    """Exact task reward: 1 if the text after TL;DR: starts with the topic."""
    scores = []
    for s in samples:
        if SEP not in s:
            scores.append(0.0)
            continue
        post, summary = s.split(SEP, 1)
        topic = post.strip().split()[0] if post.strip() else ""
        scores.append(1.0 if summary.strip().startswith(topic) and topic else 0.0)
    return scores
