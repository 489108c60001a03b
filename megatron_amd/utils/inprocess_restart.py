"""In-process restart of the training function.

Capability analog of reference megatron/training/inprocess_restart.py
(`inprocess_call_wrapper` around pretrain, nvidia-resiliency-ext): when a
step dies with a transient error (a flaky kernel launch, an injected fault,
a collective timeout after a peer hiccup), re-enter the training function in
the SAME process — resuming from the last checkpoint — instead of tearing
the job down and paying scheduler + init costs again.  Deterministic errors
(same failure twice in a row at the same iteration) abort.
"""

from __future__ import annotations

import logging
import time
from dataclasses import dataclass
from typing import Callable, Optional, Tuple, Type

logger = logging.getLogger(__name__)


@dataclass
class RestartConfig:
    max_restarts: int = 3
    # exception types considered restartable; anything else re-raises
    restartable: Tuple[Type[BaseException], ...] = (RuntimeError,)
    # two identical failures back-to-back => deterministic, abort
    abort_on_repeat: bool = True
    backoff_seconds: float = 0.0


def run_with_inprocess_restart(
    train_fn: Callable[[int], object],
    config: RestartConfig = RestartConfig(),
    on_restart: Optional[Callable[[int, BaseException], None]] = None,
):
    """Run `train_fn(attempt)` with restart-on-transient-failure.

    `train_fn` must be re-entrant: build everything from configuration and
    resume from its own checkpoints (our pretrain() is — model/optimizer are
    reconstructed and `load_checkpoint` restores exact state).  `on_restart`
    runs between attempts (e.g. destroy/reinit process groups, clear caches).
    """
    last_failure_sig = None
    attempt = 0
    while True:
        try:
            return train_fn(attempt)
        except config.restartable as e:
            sig = (type(e).__name__, str(e))
            if config.abort_on_repeat and sig == last_failure_sig:
                logger.error("identical failure twice - deterministic, aborting: %s", e)
                raise
            last_failure_sig = sig
            attempt += 1
            if attempt > config.max_restarts:
                logger.error("restart budget exhausted (%d)", config.max_restarts)
                raise
            logger.warning("in-process restart %d/%d after: %s",
                           attempt, config.max_restarts, e)
            if on_restart is not None:
                on_restart(attempt, e)
            if config.backoff_seconds:
                time.sleep(config.backoff_seconds)
