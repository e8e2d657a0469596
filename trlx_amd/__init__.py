"""trlx_amd — MI355X-native RLHF fine-tuning framework with the trlX API.

Compute path: PyTorch-ROCm orchestration + hand-written HIP/CDNA4 (gfx950)
kernels (trlx_amd/csrc) + RCCL collectives over xGMI.  See SURVEY.md for the
reference (CarperAI/trlx) structural map this build follows.
"""

__version__ = "0.1.0"

from .data.configs import TRLConfig  # noqa: F401
from .trlx import train  # noqa: F401


def release_graphs():
    """Destroy every captured hipGraph (decode engines + train steps) NOW, in
    a synchronized order.  Graph pools released by GC mid-way through a later
    context's allocations segfault on ROCm (mixed-suite crash); call this
    between independent training runs in one process (the test suite does)."""
    from .models.nn.generation import release_graphs as _rg
    from .trainer.base_trainer import release_train_graphs as _rt

    _rg()
    _rt()
