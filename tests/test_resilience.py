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
