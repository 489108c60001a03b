"""Fault injection + in-process restart (CPU)."""

import pytest
import torch

from megatron_amd.utils.fault_injection import FaultInjector, FaultInjectorConfig, InjectedFault
from megatron_amd.utils.inprocess_restart import RestartConfig, run_with_inprocess_restart


def test_fault_injector_deterministic_crash():
    inj = FaultInjector(FaultInjectorConfig(enabled=True, fault_type="crash", at_iteration=3))
    inj.maybe_inject(1)
    inj.maybe_inject(2)
    with pytest.raises(InjectedFault, match="iteration 3"):
        inj.maybe_inject(3)
    # fires once only
    inj.maybe_inject(3)


def test_fault_injector_rank_filter_and_nan():
    cfg = FaultInjectorConfig(enabled=True, fault_type="nan_loss", at_iteration=1, ranks=[2])
    inj_r0 = FaultInjector(cfg, rank=0)
    loss = torch.tensor(1.5)
    assert torch.equal(inj_r0.maybe_inject(1, loss), loss)  # rank 0 not targeted
    inj_r2 = FaultInjector(cfg, rank=2)
    assert torch.isnan(inj_r2.maybe_inject(1, loss))


def test_fault_injector_stochastic_fires_eventually():
    inj = FaultInjector(FaultInjectorConfig(enabled=True, fault_type="crash",
                                            mtti_iterations=5.0, seed=7))
    fired_at = None
    for it in range(1, 200):
        try:
            inj.maybe_inject(it)
        except InjectedFault:
            fired_at = it
            break
    assert fired_at is not None


def test_inprocess_restart_recovers_transient():
    state = {"attempts": 0, "progress": 0}

    def train_fn(attempt):
        state["attempts"] += 1
        # resume from "checkpoint": progress persists across attempts
        for step in range(state["progress"], 10):
            if step == 4 and attempt == 0:
                raise RuntimeError("transient kernel fault")
            if step == 7 and attempt == 1:
                raise RuntimeError("another transient fault")
            state["progress"] = step + 1
        return state["progress"]

    restarts = []
    out = run_with_inprocess_restart(
        train_fn, RestartConfig(max_restarts=3),
        on_restart=lambda n, e: restarts.append(str(e)))
    assert out == 10
    assert state["attempts"] == 3
    assert len(restarts) == 2


def test_inprocess_restart_aborts_on_deterministic():
    def train_fn(attempt):
        raise RuntimeError("same bug every time")

    with pytest.raises(RuntimeError, match="same bug"):
        run_with_inprocess_restart(train_fn, RestartConfig(max_restarts=5))
    # (aborted after 2 identical failures, not 5 — abort_on_repeat)


def test_inprocess_restart_budget():
    calls = []

    def train_fn(attempt):
        calls.append(attempt)
        raise RuntimeError(f"fault #{attempt}")  # distinct messages

    with pytest.raises(RuntimeError):
        run_with_inprocess_restart(train_fn, RestartConfig(max_restarts=2))
    assert len(calls) == 3  # initial + 2 restarts


def test_non_restartable_exception_propagates():
    def train_fn(attempt):
        raise ValueError("config error")

    with pytest.raises(ValueError):
        run_with_inprocess_restart(train_fn, RestartConfig())


def _rerun_dp2_case(rank, world):
    """Only rank 1 flags a bad value: BOTH ranks must replay (collective
    decision) and both must receive the same transient exit code."""
    import torch.distributed as dist

    from megatron_amd.utils.rerun_state_machine import (
        RerunDataIterator,
        RerunMode,
        RerunStateMachine,
    )

    sm = RerunStateMachine(RerunMode.VALIDATE_RESULTS)
    it = RerunDataIterator(iter([{"x": torch.tensor([float(i + rank)])} for i in range(4)]))
    runs = []
    call_count = {"n": 0}

    def flaky_value():
        # rank 1's first evaluation is bad; its replay is clean -> transient
        call_count["n"] += 1
        if rank == 1 and call_count["n"] == 1:
            return torch.tensor(float("nan"))
        return torch.tensor(1.0)

    while sm.should_run_forward_backward(it):
        batch = next(it)
        runs.append(float(batch["x"]))
        v = flaky_value()
        sm.validate_result(v, lambda t: bool(torch.isnan(t).any()), "nan loss")
    code = sm.should_checkpoint_and_exit()
    assert len(runs) == 2, runs            # every rank replayed
    assert runs[0] == runs[1]              # identical data on replay
    assert code == 17, (rank, code)        # transient verdict agreed by all


def test_rerun_collective_decision_dp2():
    from tests.utils import spawn_dist

    spawn_dist(_rerun_dp2_case, 2)


def test_rerun_state_dict_roundtrip():
    from megatron_amd.utils.rerun_state_machine import (
        RerunDataIterator,
        RerunMode,
        RerunStateMachine,
    )

    sm = RerunStateMachine(RerunMode.VALIDATE_RESULTS)
    sm.stats["transient"] = 3
    sm._step = 7
    sm2 = RerunStateMachine(RerunMode.DISABLED)
    sm2.load_state_dict(sm.state_dict())
    assert sm2.mode == RerunMode.VALIDATE_RESULTS
    assert sm2.stats["transient"] == 3 and sm2._step == 7

    it = RerunDataIterator(iter([1, 2, 3]))
    it.start_iteration()
    assert next(it) == 1
    sd = it.state_dict()
    it2 = RerunDataIterator(iter([4, 5]))
    it2.load_state_dict(sd)
    it2.rewind()
    assert next(it2) == 1  # replays the checkpointed in-flight batch
