"""Typed configuration system.

Parity target: reference trlx/data/configs.py (TRLConfig and its sub-configs,
including ``load_yaml`` / ``to_dict`` / ``from_dict`` / ``evolve`` / ``update``
dot-path merging used by sweeps).  Re-designed for the MI355X runtime: the
``TrainConfig`` carries native-runtime knobs (parallelism sizes, HIP stream
overlap) instead of accelerate/DeepSpeed plumbing.
"""

from copy import deepcopy
from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional, Set

import yaml

from .method_configs import MethodConfig, get_method


def merge(base: Dict, update: Dict, updated: Set[str]) -> Dict:
    """Recursively merge ``update`` into ``base``, recording merged keys."""
    for k, v in base.items():
        if k in update and isinstance(v, dict):
            base[k] = merge(v, update[k], updated)
            updated.add(k)
        elif k in update:
            base[k] = update[k]
            updated.add(k)
    return base


def _merge_dicts(base: Dict, update: Dict) -> Dict:
    """Merge ``update`` into ``base``, raising on unknown keys."""
    for k in update:
        if k not in base:
            raise ValueError(f"Unknown config key: {k}")
    out = deepcopy(base)
    updated: Set[str] = set()
    merge(out, update, updated)
    return out


@dataclass
class ModelConfig:
    """Which model to fine-tune and how much of it to train.

    :param model_path: HF-style directory (or registry name) of the base model.
    :param model_arch_type: "causal" or "seq2seq".
    :param num_layers_unfrozen: number of top decoder layers left trainable;
        -1 trains everything.  The frozen bottom + the hydra reference branch
        share storage in HBM (288 GB/GPU allows a resident reference).
    :param peft_config: optional dict/peft config for parameter-efficient
        fine-tuning (LoRA etc.).
    """

    model_path: str
    model_arch_type: str = "causal"
    num_layers_unfrozen: int = -1
    model_extra_configs: Dict[str, Any] = field(default_factory=dict)
    peft_config: Optional[Any] = None
    # recompute block activations in backward (parity: NeMo
    # activations_checkpoint_method, megatron_20b.yaml:77-79) — trades ~30%
    # step time for activation memory on long-context runs
    gradient_checkpointing: bool = False
    # K15: keep the hydra reference branch's block weights in pinned host
    # memory and stream them to the GPU just in time (parity: NeMo
    # offload_reference_model, megatron_65b.yaml:5-6).  Only worth it on
    # 65B-class replicas; 288 GB HBM keeps <=20B references resident.
    ref_offload: bool = False

    @classmethod
    def from_dict(cls, config: Dict[str, Any]) -> "ModelConfig":
        return cls(**config)


@dataclass
class TokenizerConfig:
    """Tokenizer selection and padding behaviour."""

    tokenizer_path: str
    padding_side: str = "left"
    truncation_side: str = "right"
    tokenizer_extra_configs: Dict[str, Any] = field(default_factory=dict)

    @classmethod
    def from_dict(cls, config: Dict[str, Any]) -> "TokenizerConfig":
        return cls(**config)


@dataclass
class OptimizerConfig:
    """Optimizer selection (name registry mirrors the reference's)."""

    name: str = "adamw"
    kwargs: Dict[str, Any] = field(default_factory=dict)

    @classmethod
    def from_dict(cls, config: Dict[str, Any]) -> "OptimizerConfig":
        return cls(**config)


@dataclass
class SchedulerConfig:
    """LR scheduler selection."""

    name: str = "cosine_annealing"
    kwargs: Dict[str, Any] = field(default_factory=dict)

    @classmethod
    def from_dict(cls, config: Dict[str, Any]) -> "SchedulerConfig":
        return cls(**config)


@dataclass
class TrainConfig:
    """Training loop + native-runtime configuration.

    The reference's accelerate knobs (``trackers``, intervals, checkpointing,
    minibatching) are kept 1:1; the distributed section replaces
    accelerate/DeepSpeed with the native runtime's explicit parallelism sizes.
    """

    total_steps: int = 10000
    seq_length: int = 1024
    epochs: int = 100
    batch_size: int = 32
    minibatch_size: Optional[int] = None

    checkpoint_interval: int = 1000
    eval_interval: int = 100

    pipeline: str = "PromptPipeline"
    trainer: str = "PPOTrainer"
    trainer_kwargs: Dict[str, Any] = field(default_factory=dict)

    project_name: str = "trlx_amd"
    entity_name: Optional[str] = None
    group_name: Optional[str] = None
    run_name: Optional[str] = None

    checkpoint_dir: str = "ckpts"
    rollout_logging_dir: Optional[str] = None
    save_best: bool = True
    save_optimizer: bool = True

    tracker: Optional[str] = "tensorboard"
    logging_dir: Optional[str] = None
    tags: List[str] = field(default_factory=list)

    seed: int = 1000

    resume_from_checkpoint: Optional[str] = None

    # --- native runtime ---------------------------------------------------
    # compute dtype for forward/backward ("bf16" | "fp32"); master weights
    # are always fp32 inside the fused Adam.
    mixed_precision: str = "bf16"
    # tensor-parallel / pipeline-parallel degrees; data-parallel size is
    # derived as world_size // (tp * pp).
    tensor_parallel_size: int = 1
    pipeline_parallel_size: int = 1
    sequence_parallel: bool = False
    # ZeRO-style optimizer-state sharding over the DP group (0 = replicated).
    zero_stage: int = 0
    # gradient all-reduce bucket size in MB (sized for xGMI: 7x153 GB/s links)
    bucket_size_mb: int = 128
    # capture the inner training step in a hipGraph when launch-bound
    use_hip_graphs: bool = False
    activation_checkpointing: bool = False

    @classmethod
    def from_dict(cls, config: Dict[str, Any]) -> "TrainConfig":
        return cls(**config)


@dataclass
class TRLConfig:
    """Top-level config — the single object handed to every trainer."""

    method: MethodConfig
    model: ModelConfig
    optimizer: OptimizerConfig
    scheduler: SchedulerConfig
    tokenizer: TokenizerConfig
    train: TrainConfig

    @classmethod
    def load_yaml(cls, yml_fp: str) -> "TRLConfig":
        with open(yml_fp, mode="r") as file:
            config = yaml.safe_load(file)
        return cls.from_dict(config)

    def to_dict(self) -> Dict[str, Any]:
        data = {
            "method": {**self.method.__dict__, "name": type(self.method).__name__},
            "model": self.model.__dict__.copy(),
            "optimizer": self.optimizer.__dict__.copy(),
            "scheduler": self.scheduler.__dict__.copy(),
            "tokenizer": self.tokenizer.__dict__.copy(),
            "train": self.train.__dict__.copy(),
        }
        return deepcopy(data)

    def evolve(self, **kwargs) -> "TRLConfig":
        """Return a new config with nested-dict overrides merged in."""
        return TRLConfig.from_dict(_merge_dicts(self.to_dict(), kwargs))

    @classmethod
    def from_dict(cls, config: Dict[str, Any]) -> "TRLConfig":
        config = deepcopy(config)
        method = config["method"]
        if isinstance(method, MethodConfig):
            pass
        else:
            method = get_method(method["name"]).from_dict(method)
        return cls(
            method=method,
            model=ModelConfig.from_dict(config["model"]),
            optimizer=OptimizerConfig.from_dict(config["optimizer"]),
            scheduler=SchedulerConfig.from_dict(config["scheduler"]),
            tokenizer=TokenizerConfig.from_dict(config["tokenizer"]),
            train=TrainConfig.from_dict(config["train"]),
        )

    @classmethod
    def update(cls, baseconfig: Dict, config: Dict) -> "TRLConfig":
        """Dot-path override merge (the sweep/CLI-hparams channel).

        ``{"method.lr": 1e-4}`` or nested dicts both work, mirroring the
        reference's ``TRLConfig.update``.
        """
        update = {}
        # unflatten dot-paths (dict values are allowed at any depth and merge
        # recursively — a superset of the reference's top-level-only dicts)
        for name, value in config.items():
            *layers, var = name.split(".")
            d = update
            for layer in layers:
                d = d.setdefault(layer, {})
            d[var] = value

        if not isinstance(baseconfig, dict):
            baseconfig = baseconfig.to_dict()

        updates: Set[str] = set()
        merged = merge(deepcopy(baseconfig), update, updates)
        for param in update:
            if param not in updates:
                raise ValueError(f"parameter {param} is not present in the config")

        return cls.from_dict(merged)

    def __str__(self) -> str:
        import json

        return json.dumps(self.to_dict(), indent=2, default=str)
