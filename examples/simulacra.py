"""Simulacra (parity: reference examples/simulacra.py — ILQL on image-prompt
/ aesthetic-rating pairs from simulacra-aesthetic-captions).

Offline adaptation: the sqlite download is unavailable, so a synthetic
prompt/rating corpus with the same shape (caption strings, 1-10 ratings
correlated with "aesthetic" phrasing) stands in; the training call is the
same offline-ILQL `trlx.train(samples=..., rewards=...)`.
"""

import json
import random
import sys

import os as _os
import sys as _sys
_sys.path.insert(0, _os.path.dirname(_os.path.dirname(_os.path.abspath(__file__))))
import trlx_amd as trlx
from trlx_amd.data.default_configs import default_ilql_config
from trlx_amd.models.nn.config import preset

SUBJECTS = ["an astronaut", "a castle", "a forest", "a city street", "a dragon",
            "an ocean wave", "a mountain peak", "a lighthouse"]
STYLES = ["oil painting", "watercolor", "pencil sketch", "digital art",
          "photograph", "charcoal drawing"]
GOOD = ["highly detailed", "masterpiece", "trending", "dramatic lighting", "4k"]
PLAIN = ["low effort", "blurry", "rough", "plain", "draft"]


def synthetic_corpus(n=512, seed=0):
    rng = random.Random(seed)
    samples, ratings = [], []
    for _ in range(n):
        subject = rng.choice(SUBJECTS)
        style = rng.choice(STYLES)
        good = rng.random() < 0.5
        tag = rng.choice(GOOD if good else PLAIN)
        samples.append(f"{subject} as a {style}, {tag}")
        base = 7.0 if good else 3.0
        ratings.append(max(1.0, min(10.0, rng.gauss(base, 1.0))))
    return samples, ratings


def default_config():
    config = default_ilql_config()
    config.model.model_path = "gpt2"
    config.model.model_extra_configs = {"config": preset("gpt2").to_dict()}
    config.tokenizer.tokenizer_path = "byte"
    config.train.seq_length = 64
    config.train.batch_size = 32
    return config


def main(hparams={}):
    config = trlx.TRLConfig.update(default_config().to_dict(), hparams)
    samples, ratings = synthetic_corpus()
    trlx.train(
        samples=samples,
        rewards=ratings,
        eval_prompts=["an astronaut riding a horse"] * 8,
        config=config,
    )


if __name__ == "__main__":
    main({} if len(sys.argv) == 1 else json.loads(sys.argv[1]))
