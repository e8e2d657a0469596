"""Trainer registry + abstract base.

Parity target: reference trlx/trainer/__init__.py (``_TRAINERS``,
``@register_trainer``, BaseRLTrainer).
"""

import sys
from abc import abstractmethod
from typing import Any, Callable, Dict, Iterable, Optional

from ..data.configs import TRLConfig
from ..pipeline import BaseRolloutStore

_TRAINERS: Dict[str, type] = {}


def register_trainer(name):
    """Decorator registering a trainer class by name."""

    def register_class(cls, name):
        _TRAINERS[name] = cls
        setattr(sys.modules[__name__], name, cls)
        return cls

    if isinstance(name, str):
        name = name.lower()
        return lambda c: register_class(c, name)

    cls = name
    name = cls.__name__
    register_class(cls, name.lower())
    return cls


@register_trainer
class BaseRLTrainer:
    def __init__(
        self,
        config: TRLConfig,
        reward_fn=None,
        metric_fn=None,
        logit_mask=None,
        stop_sequences=None,
        train_mode: bool = True,
    ):
        self.store: Optional[BaseRolloutStore] = None
        self.config = config
        self.reward_fn = reward_fn
        self.metric_fn = metric_fn
        self.train_mode = train_mode
        self.logit_mask = logit_mask
        self.stop_sequences = stop_sequences

    def push_to_store(self, data):
        self.store.push(data)

    def add_eval_pipeline(self, eval_pipeline):
        """Adds a evaluation pipeline with validation prompts"""
        self.eval_pipeline = eval_pipeline

    @abstractmethod
    def learn(self):
        """Train the model and log the metrics."""
        pass

    @abstractmethod
    def save(self, directory: Optional[str] = None):
        """Save the checkpoint of the current state."""
        pass

    @abstractmethod
    def load(self, directory: Optional[str] = None):
        """Load checkpoint of the current state."""
        pass
