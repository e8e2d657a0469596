from .config import TransformerConfig, preset  # noqa: F401
from .transformer import CausalTransformer  # noqa: F401
