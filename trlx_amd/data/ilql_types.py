"""ILQL data types + dataclass<->tensor-list transport helpers.

Parity target: reference trlx/data/ilql_types.py (ILQLElement / ILQLBatch and
their Seq2Seq variants).  The reference *imports* ``flatten_dataclass`` /
``unflatten_dataclass`` from this module in four places but never defines them
 — they are defined for real here, since pipeline-parallel
transport needs dataclass ⇄ ordered-tensor-list conversion.
"""

from dataclasses import dataclass, fields
from typing import Callable, List, Type

from torch import Tensor


def flatten_dataclass(cls: Type) -> Callable:
    """Return a function mapping an instance of ``cls`` to a list of its
    field values in declaration order (tensor transport order)."""
    cls_fields = [f.name for f in fields(cls)]

    def flatten(x) -> List:
        return [getattr(x, name) for name in cls_fields]

    return flatten


def unflatten_dataclass(cls: Type) -> Callable:
    """Return a function mapping an ordered value list back to ``cls``."""
    cls_fields = [f.name for f in fields(cls)]

    def unflatten(values: List):
        return cls(**dict(zip(cls_fields, values)))

    return unflatten


@dataclass
class ILQLElement:
    """One offline RL training element.

    :param input_ids: full dialogue tokens ``[T]``
    :param attention_mask: ``[T]``
    :param rewards: per-action rewards ``[A]``
    :param states_ixs: indices of state positions ``[A+1]``
    :param actions_ixs: indices of action (output token) positions ``[A]``
    :param dones: terminal flags per state ``[A+1]``
    """

    input_ids: Tensor
    attention_mask: Tensor
    rewards: Tensor
    states_ixs: Tensor
    actions_ixs: Tensor
    dones: Tensor


@dataclass
class ILQLSeq2SeqElement:
    """Seq2seq variant carrying separate decoder input."""

    input_ids: Tensor
    attention_mask: Tensor
    decoder_input_ids: Tensor
    rewards: Tensor
    states_ixs: Tensor
    actions_ixs: Tensor
    dones: Tensor


@dataclass
class ILQLBatch:
    """Batched ILQL data (padded)."""

    input_ids: Tensor
    attention_mask: Tensor
    rewards: Tensor
    states_ixs: Tensor
    actions_ixs: Tensor
    dones: Tensor


@dataclass
class ILQLSeq2SeqBatch:
    """Batched seq2seq ILQL data (padded)."""

    input_ids: Tensor
    attention_mask: Tensor
    decoder_input_ids: Tensor
    rewards: Tensor
    states_ixs: Tensor
    actions_ixs: Tensor
    dones: Tensor
