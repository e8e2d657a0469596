"""Synthetic Helpful-Harmless dialogue task (offline stand-in for
Anthropic/hh-rlhf used by reference examples/hh/ppo_hh.py — no network).

Dialogues follow the HH format ("\n\nHuman: ...\n\nAssistant: ...").  The
oracle prefers responses containing HELPFUL words and penalizes RUDE words,
so PPO has a recoverable signal and eval has an exact metric.
"""

import random
from typing import List

HELPFUL = ["sure", "help", "glad", "here", "can", "yes", "of", "course"]
RUDE = ["no", "never", "go", "away", "busy", "wrong"]
QUESTIONS = [
    "how do I bake bread", "what is the capital of France", "fix my bike",
    "explain photosynthesis", "tips for sleeping well", "how to learn piano",
    "what causes rain", "recommend a book", "how do magnets work",
]

PROMPT_FMT = "\n\nHuman: {q}\n\nAssistant:"


def make_prompts(n: int, seed: int = 0) -> List[str]:
    rng = random.Random(seed)
    return [PROMPT_FMT.format(q=rng.choice(QUESTIONS)) for _ in range(n)]


def make_pairs(n: int, seed: int = 1):
    """(chosen, rejected) dialogue pairs for reward-model training."""
    rng = random.Random(seed)
    pairs = []
    for _ in range(n):
        p = PROMPT_FMT.format(q=rng.choice(QUESTIONS))
        good = " " + " ".join(rng.choices(HELPFUL, k=rng.randint(3, 6)))
        bad = " " + " ".join(rng.choices(RUDE, k=rng.randint(2, 5)))
        pairs.append((p + good, p + bad))
    return pairs


def oracle_reward(samples: List[str]) -> List[float]:
    scores = []
    for s in samples:
        resp = s.split("Assistant:")[-1].lower()
        words = resp.split()
        if not words:
            scores.append(0.0)
            continue
        pos = sum(w.strip(".,!?") in HELPFUL for w in words)
        neg = sum(w.strip(".,!?") in RUDE for w in words)
        scores.append((pos - neg) / max(len(words), 1))
    return scores
