"""Prompt / generic RL element batch types.

Parity target: reference trlx/data/accelerate_base_datatypes.py
(PromptElement/PromptBatch, AccelerateRLElement/Batch) — renamed without the
accelerate prefix since the runtime is native.
"""

from dataclasses import dataclass
from typing import Iterable

from torch import Tensor


@dataclass
class PromptElement:
    """A single tokenized prompt."""

    text: str
    tokens: Tensor


@dataclass
class PromptBatch:
    """A batch of tokenized prompts."""

    text: Iterable[str]
    tokens: Tensor


@dataclass
class RLElement:
    """Generic output tokens + per-token rewards."""

    output_tokens: Tensor
    rewards: Tensor


@dataclass
class RLBatchElement:
    """Batched output tokens + rewards."""

    output_tokens: Tensor
    rewards: Tensor
