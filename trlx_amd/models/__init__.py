from .modeling_base import PreTrainedModelWrapper  # noqa: F401
from .nn.config import TransformerConfig, preset  # noqa: F401
from .nn.transformer import CausalTransformer  # noqa: F401
