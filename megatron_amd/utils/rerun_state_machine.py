"""Rerun state machine: attribute NaN/spiky losses to transient hardware
faults vs deterministic bugs by replaying the same iteration.

Capability analog of reference megatron/core/rerun_state_machine.py (:60ff
RerunStateMachine, validate_result API, RerunDataIterator, exit codes :37-40):
the train loop asks ``should_run_forward_backward()`` in a while loop; when a
``validate_result`` call flags an unexpected value the machine requests one
replay of the identical iteration (data replayed from the iterator wrapper):

* replay reproduces the bad value  -> deterministic (likely software) error;
  EXIT_CODE_FAILED_ON_RESULT_VALIDATION (16).
* replay differs                   -> transient hardware fault; logged, and
  the machine can request checkpoint-and-exit (17) so the job restarts clean.
"""

from __future__ import annotations

import logging
from enum import Enum
from typing import Any, Callable, List, Optional

import torch

logger = logging.getLogger(__name__)

EXIT_CODE_FAILED_ON_RESULT_VALIDATION = 16
EXIT_CODE_SUCCESS_ON_RESULT_VALIDATION = 17  # transient: save+exit requested


class RerunMode(str, Enum):
    DISABLED = "disabled"
    VALIDATE_RESULTS = "validate_results"


class RerunState(Enum):
    NOT_RUNNING_YET = 0
    FIRST_RUN = 1
    RERUNNING = 2


class RerunDataIterator:
    """Wraps a data iterator so an iteration's batches can be replayed."""

    def __init__(self, iterable):
        self._it = iterable
        self._history: List[Any] = []
        self._replaying = False
        self._replay_pos = 0

    def __iter__(self):
        return self

    def __next__(self):
        if self._replaying:
            if self._replay_pos >= len(self._history):
                raise StopIteration
            out = self._history[self._replay_pos]
            self._replay_pos += 1
            return out
        batch = next(self._it)
        self._history.append(batch)
        return batch

    def state_dict(self) -> dict:
        """Checkpointable replay state (reference RerunDataIterator): lets a
        restarted job replay the in-flight iteration's batches."""
        return {"history": self._history, "replaying": self._replaying,
                "replay_pos": self._replay_pos}

    def load_state_dict(self, sd: dict):
        self._history = list(sd.get("history", []))
        self._replaying = bool(sd.get("replaying", False))
        self._replay_pos = int(sd.get("replay_pos", 0))

    def start_iteration(self):
        self._history.clear()
        self._replaying = False

    def rewind(self):
        self._replaying = True
        self._replay_pos = 0


class RerunStateMachine:
    """State machine driving at-most-one replay per flagged iteration."""

    def __init__(self, mode: RerunMode = RerunMode.DISABLED,
                 error_injection_rate: float = 0.0):
        self._local_request = False
        self.mode = RerunMode(mode)
        self.state = RerunState.NOT_RUNNING_YET
        self._first_values: List[float] = []
        self._rerun_values: List[float] = []
        self._failed_msg: Optional[str] = None
        self._request_rerun = False
        self.stats = {"reruns": 0, "transient": 0, "persistent": 0}
        self._value_index = 0
        self._error_injection_rate = error_injection_rate
        self._step = 0

    # -- train-loop protocol ---------------------------------------------

    def should_run_forward_backward(self, data_iterators) -> bool:
        """Call in a while-loop around the forward/backward. Returns True for
        the first run and once more when a replay is requested."""
        its = data_iterators if isinstance(data_iterators, (list, tuple)) else [data_iterators]
        if self.state in (RerunState.NOT_RUNNING_YET,):
            self.state = RerunState.FIRST_RUN
            self._first_values.clear()
            self._failed_msg = None
            self._request_rerun = False
            self._local_request = False
            self._value_index = 0
            self._step += 1
            for it in its:
                if isinstance(it, RerunDataIterator):
                    it.start_iteration()
            return True
        if self.state == RerunState.FIRST_RUN:
            # COLLECTIVE decision: if any rank flagged a result, every rank
            # must replay, or the replayed iteration's collectives desync
            # and the job hangs (reference rerun_state_machine rank sync)
            self._local_request = self._request_rerun
            if torch.distributed.is_initialized() and torch.distributed.get_world_size() > 1:
                flag = torch.tensor([1 if self._request_rerun else 0])
                torch.distributed.all_reduce(flag, op=torch.distributed.ReduceOp.MAX)
                self._request_rerun = bool(int(flag))
        if self.state == RerunState.FIRST_RUN and self._request_rerun:
            self.state = RerunState.RERUNNING
            self._rerun_values.clear()
            self._value_index = 0
            self.stats["reruns"] += 1
            for it in its:
                if isinstance(it, RerunDataIterator):
                    it.rewind()
            return True
        return False

    def should_checkpoint_and_exit(self) -> Optional[int]:
        """After the while-loop: non-None exit code if the machine concluded
        this iteration hit a fault.  The verdict comes from the rank(s) that
        DETECTED the bad value (non-flagging ranks trivially reproduce their
        own healthy values) and is agreed collectively so every rank exits
        with the same code."""
        if self.state == RerunState.RERUNNING:
            local_transient = 0
            local_persistent = 0
            if self._local_request:
                same = len(self._first_values) == len(self._rerun_values) and all(
                    (a == b) or (a != a and b != b)  # NaN == NaN for this purpose
                    for a, b in zip(self._first_values, self._rerun_values))
                if same:
                    local_persistent = 1
                    logger.error("rerun reproduced the invalid result: deterministic error (%s)",
                                 self._failed_msg)
                else:
                    local_transient = 1
                    logger.error("rerun produced a different result: transient fault (%s)",
                                 self._failed_msg)
            if torch.distributed.is_initialized() and torch.distributed.get_world_size() > 1:
                flags = torch.tensor([local_transient, local_persistent])
                torch.distributed.all_reduce(flags, op=torch.distributed.ReduceOp.MAX)
                local_transient, local_persistent = int(flags[0]), int(flags[1])
            self.state = RerunState.NOT_RUNNING_YET
            if local_persistent:
                self.stats["persistent"] += 1
                return EXIT_CODE_FAILED_ON_RESULT_VALIDATION
            if local_transient:
                self.stats["transient"] += 1
                return EXIT_CODE_SUCCESS_ON_RESULT_VALIDATION
            return None
        self.state = RerunState.NOT_RUNNING_YET
        return None

    # -- checkpointing (reference RerunStateMachine state_dict) -------------

    def state_dict(self) -> dict:
        return {"mode": self.mode.value, "stats": dict(self.stats), "step": self._step}

    def load_state_dict(self, sd: dict):
        self.mode = RerunMode(sd.get("mode", self.mode.value))
        self.stats.update(sd.get("stats", {}))
        self._step = sd.get("step", self._step)

    # -- called from loss functions ---------------------------------------

    def validate_result(self, result: torch.Tensor, rejection_func: Callable[[torch.Tensor], bool],
                        message: str = "unexpected result") -> None:
        if self.mode == RerunMode.DISABLED:
            return
        val = float(result.detach().float().sum()) if torch.is_tensor(result) else float(result)
        inject = False
        if self._error_injection_rate > 0 and self.state == RerunState.FIRST_RUN:
            # deterministic fault injection for tests: flag every Nth step
            inject = (self._step % max(int(1 / self._error_injection_rate), 1)) == 0
        bad = bool(rejection_func(result)) or inject
        if self.state == RerunState.FIRST_RUN:
            self._first_values.append(val)
            if bad:
                self._request_rerun = True
                self._failed_msg = message
        elif self.state == RerunState.RERUNNING:
            self._rerun_values.append(val)
        self._value_index += 1


_GLOBAL: RerunStateMachine = RerunStateMachine()


def get_rerun_state_machine() -> RerunStateMachine:
    return _GLOBAL


def initialize_rerun_state_machine(mode: str = "disabled", **kw) -> RerunStateMachine:
    global _GLOBAL
    _GLOBAL = RerunStateMachine(RerunMode(mode), **kw)
    return _GLOBAL
