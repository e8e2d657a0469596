"""Registry loading (parity: reference trlx/utils/loading.py)."""

# isort: off
from ..pipeline import _DATAPIPELINE
from ..pipeline.offline_pipeline import PromptPipeline  # noqa: F401 (registers)
from ..trainer import _TRAINERS
from ..trainer.ppo_trainer import PPOTrainer  # noqa: F401 (registers)
from ..trainer.ilql_trainer import ILQLTrainer  # noqa: F401
from ..trainer.sft_trainer import SFTTrainer  # noqa: F401
from ..trainer.rft_trainer import RFTTrainer  # noqa: F401

# isort: on

# aliases matching the reference's trainer names so existing configs load
_ALIASES = {
    "accelerateppotrainer": "ppotrainer",
    "accelerateilqltrainer": "ilqltrainer",
    "acceleratesfttrainer": "sfttrainer",
    "acceleraterfttrainer": "rfttrainer",
    "nemoppotrainer": "ppotrainer",
    "nemoilqltrainer": "ilqltrainer",
    "nemosfttrainer": "sfttrainer",
}


def get_trainer(name: str) -> type:
    """Return the trainer class registered under ``name``."""
    name = name.lower()
    name = _ALIASES.get(name, name)
    if name in _TRAINERS:
        return _TRAINERS[name]
    raise Exception(f"Error: Trying to access a trainer that has not been registered: {name}")


def get_pipeline(name: str) -> type:
    """Return the pipeline class registered under ``name``."""
    name = name.lower()
    if name in _DATAPIPELINE:
        return _DATAPIPELINE[name]
    raise Exception(f"Error: Trying to access a pipeline that has not been registered: {name}")
