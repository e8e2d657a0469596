"""trlx_amd — MI355X-native RLHF fine-tuning framework with the trlX API.

Compute path: PyTorch-ROCm orchestration + hand-written HIP/CDNA4 (gfx950)
kernels (trlx_amd/csrc) + RCCL collectives over xGMI.  See SURVEY.md for the
reference (CarperAI/trlx) structural map this build follows.
"""

__version__ = "0.1.0"

from .data.configs import TRLConfig  # noqa: F401
from .trlx import train  # noqa: F401
