"""Metric trackers (reference uses accelerate's wandb/tensorboard plumbing;
wandb is unavailable offline, so: tensorboard when importable, a JSONL file
tracker otherwise/always)."""

import json
import os
import time
from typing import Dict, Optional

from . import filter_non_scalars


class JsonlTracker:
    """Appends one JSON object per logged step to <dir>/metrics.jsonl."""

    def __init__(self, logging_dir: str, config: Optional[dict] = None):
        os.makedirs(logging_dir, exist_ok=True)
        self.path = os.path.join(logging_dir, "metrics.jsonl")
        self._f = open(self.path, "a")
        if config is not None:
            with open(os.path.join(logging_dir, "config.json"), "w") as f:
                json.dump(config, f, indent=2, default=str)

    def log(self, stats: Dict, step: int):
        rec = {"step": step, "time": time.time(), **filter_non_scalars(stats)}
        self._f.write(json.dumps(rec) + "\n")
        self._f.flush()

    def finish(self):
        self._f.close()


class TensorboardTracker:
    def __init__(self, logging_dir: str, config: Optional[dict] = None):
        from torch.utils.tensorboard import SummaryWriter

        self.writer = SummaryWriter(logging_dir)
        if config is not None:
            self.writer.add_text("config", json.dumps(config, indent=2, default=str))

    def log(self, stats: Dict, step: int):
        for k, v in filter_non_scalars(stats).items():
            self.writer.add_scalar(k, v, step)

    def finish(self):
        self.writer.close()


class NoopTracker:
    def log(self, stats: Dict, step: int):
        pass

    def finish(self):
        pass


def make_tracker(name: Optional[str], logging_dir: str, config: Optional[dict] = None,
                 main_process: bool = True):
    """Build the tracker for this run (rank 0 only; others get a no-op)."""
    if not main_process or name is None:
        return NoopTracker()
    if name == "tensorboard":
        try:
            return TensorboardTracker(logging_dir, config)
        except ImportError:
            return JsonlTracker(logging_dir, config)
    if name in ("jsonl", "wandb"):  # wandb unavailable offline -> jsonl
        return JsonlTracker(logging_dir, config)
    raise ValueError(f"Unknown tracker: {name} (supported: tensorboard, jsonl, None)")
