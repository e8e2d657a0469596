"""A grounded list-manipulation DSL (parity: the reference's
examples/experiments/grounded_program_synthesis — a toy language whose
programs transform integer lists; the model learns to emit a program that
maps a given input to a given output, graded by actually RUNNING it).

This is an independent, compact implementation of the same idea: a registry
of primitives, a safe recursive-descent interpreter (no eval of model
output), a random program/dataset generator, and an exact-match reward.
"""

import random
from typing import List, Optional

MAX_INT = 8


def _clamp(xs):
    return [max(-MAX_INT, min(MAX_INT, x)) for x in xs]


PRIMITIVES = {
    "reverse": lambda xs: xs[::-1],
    "sort_asc": lambda xs: sorted(xs),
    "sort_desc": lambda xs: sorted(xs, reverse=True),
    "drop_first": lambda xs: xs[1:],
    "drop_last": lambda xs: xs[:-1],
    "halve": lambda xs: xs[: max(1, len(xs) // 2)],
    "neg": lambda xs: _clamp([-x for x in xs]),
    "inc": lambda xs: _clamp([x + 1 for x in xs]),
    "dec": lambda xs: _clamp([x - 1 for x in xs]),
    "double": lambda xs: _clamp([2 * x for x in xs]),
}


def run_program(program: str, xs: List[int]) -> Optional[List[int]]:
    """Interpret ``f;g;h`` as h(g(f(xs))); None on any malformed program."""
    if not program.strip():
        return None
    out = list(xs)
    for name in program.strip().split(";"):
        fn = PRIMITIVES.get(name.strip())
        if fn is None:
            return None
        out = fn(out)
        if not out:
            return None
    return out


def random_program(rng: random.Random, max_len: int = 3) -> str:
    n = rng.randint(1, max_len)
    return ";".join(rng.choice(sorted(PRIMITIVES)) for _ in range(n))


def random_input(rng: random.Random) -> List[int]:
    return [rng.randint(-MAX_INT, MAX_INT) for _ in range(rng.randint(3, 6))]


def format_prompt(xs: List[int], ys: List[int]) -> str:
    return f"Input: {xs} Output: {ys} Function:"


def parse_sample(sample: str):
    """Extract (input, output, program) from a generated sample string."""
    try:
        inp = sample.split("Input:")[1].split("Output:")[0].strip()
        out = sample.split("Output:")[1].split("Function:")[0].strip()
        prog = sample.split("Function:")[1].strip().split()[0] if \
            sample.split("Function:")[1].strip() else ""
        xs = [int(t) for t in inp.strip("[]").split(",") if t.strip()]
        ys = [int(t) for t in out.strip("[]").split(",") if t.strip()]
        return xs, ys, prog
    except (IndexError, ValueError):
        return None


def make_dataset(n: int = 512, seed: int = 0):
    """(prompt, gold_program) pairs whose outputs are grounded by execution."""
    rng = random.Random(seed)
    data = []
    while len(data) < n:
        xs = random_input(rng)
        prog = random_program(rng)
        ys = run_program(prog, xs)
        if ys is None:
            continue
        data.append((format_prompt(xs, ys), prog))
    return data


def reward_fn(samples, **kwargs):
    """+1 exact output match, -0.5 runnable but wrong, -1 unparsable
    (the reference's grading scheme, grounded by interpretation)."""
    rewards = []
    for s in samples:
        parsed = parse_sample(s)
        if parsed is None:
            rewards.append(-1.0)
            continue
        xs, ys, prog = parsed
        got = run_program(prog, xs)
        if got is None:
            rewards.append(-1.0)
        elif got == ys:
            rewards.append(1.0)
        else:
            rewards.append(-0.5)
    return rewards
