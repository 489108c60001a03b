"""Training metrics logging.

Capability analog of the reference's training_log sinks (TensorBoard writer,
wandb, progress log — training.py:2606, :2992). The tensorboard package is not
in the image, so the durable sink is a JSONL event log (one object per logged
step: {"step": N, "ts": ..., metric: value, ...}) that tooling can tail or
convert; wandb is used when importable.
"""

from __future__ import annotations

import json
import os
import time
from typing import Optional


class MetricsLogger:
    def __init__(self, log_dir: Optional[str] = None, rank: int = 0, use_wandb: bool = False,
                 wandb_project: Optional[str] = None):
        self._file = None
        self._wandb = None
        if rank != 0:
            return
        if log_dir:
            os.makedirs(log_dir, exist_ok=True)
            self._file = open(os.path.join(log_dir, "metrics.jsonl"), "a", buffering=1)
        if use_wandb:
            try:
                import wandb

                self._wandb = wandb
                wandb.init(project=wandb_project or "megatron_amd")
            except ImportError:
                self._wandb = None

    def log(self, step: int, **metrics):
        if self._file is not None:
            self._file.write(json.dumps({"step": step, "ts": time.time(), **metrics}) + "\n")
        if self._wandb is not None:
            self._wandb.log(metrics, step=step)

    def close(self):
        if self._file is not None:
            self._file.close()
        if self._wandb is not None:
            self._wandb.finish()


def append_progress_log(save_dir: str, rank: int, message: str):
    """progress.txt append (reference training.py:2992 analog)."""
    if rank != 0 or not save_dir:
        return
    os.makedirs(save_dir, exist_ok=True)
    with open(os.path.join(save_dir, "progress.txt"), "a") as f:
        f.write(f"{time.strftime('%Y-%m-%d %H:%M:%S')}\t{message}\n")
