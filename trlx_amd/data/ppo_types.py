"""PPO rollout data types.

Parity target: reference trlx/data/ppo_types.py (PPORLElement / PPORLBatch,
all tensors per-token).
"""

from dataclasses import dataclass

from torch import Tensor


@dataclass
class PPORLElement:
    """A single rollout element.

    :param query_tensor: prompt tokens ``[Q]``
    :param response_tensor: generated tokens ``[R]``
    :param logprobs: policy log-probs per response token ``[R]``
    :param values: value-head outputs per response token ``[R]``
    :param rewards: per-token rewards (KL penalty + terminal score) ``[R]``
    """

    query_tensor: Tensor
    response_tensor: Tensor
    logprobs: Tensor
    values: Tensor
    rewards: Tensor


@dataclass
class PPORLBatch:
    """A batched, padded view of PPORLElements.

    :param query_tensors: left-padded ``[B, Q_max]``
    :param response_tensors: right-padded ``[B, R_max]``
    :param logprobs: ``[B, R_max]``
    :param values: ``[B, R_max]``
    :param rewards: ``[B, R_max]``
    """

    query_tensors: Tensor
    response_tensors: Tensor
    logprobs: Tensor
    values: Tensor
    rewards: Tensor
