"""Method-config registry.

Parity target: reference trlx/data/method_configs.py (``@register_method``,
``MethodConfig``, ``get_method``).
"""

import sys
from dataclasses import dataclass, field
from typing import Any, Callable, Dict

# registry: lowercase method name -> MethodConfig subclass
_METHODS: Dict[str, type] = {}


def register_method(name: Any = None) -> Callable:
    """Decorator used to register a method config by (class) name."""

    def register_class(cls, name):
        _METHODS[name] = cls
        setattr(sys.modules[__name__], name, cls)
        return cls

    if isinstance(name, str):
        name = name.lower()
        return lambda c: register_class(c, name)

    cls = name
    name = cls.__name__
    register_class(cls, name.lower())
    return cls


@dataclass
@register_method
class MethodConfig:
    """Base config for a training method (PPO / ILQL / SFT / RFT).

    :param name: method registry name
    """

    name: str = "MethodConfig"

    @classmethod
    def from_dict(cls, config: Dict[str, Any]) -> "MethodConfig":
        return cls(**config)


def get_method(name: str) -> type:
    """Return the MethodConfig subclass registered under ``name``."""
    name = name.lower()
    if name in _METHODS:
        return _METHODS[name]
    raise Exception(f"Error: Trying to access a method that has not been registered: {name}")
