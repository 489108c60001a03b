"""Pre-run GPU health check.

Capability analog of reference megatron/training/gpu_sniff_test.py
(hooked at training.py:3175): before committing a long job to a node, run
a deterministic bf16 GEMM on every visible GPU and compare against a CPU
fp32 reference — catching downclocked, mis-seated, or silently-corrupting
accelerators (on MI355X also a cheap way to surface an unhealthy XCD)."""

from __future__ import annotations

from typing import List, Optional

import torch


def gpu_sniff_test(size: int = 2048, tol: float = 2e-2,
                   device: Optional[torch.device] = None) -> List[str]:
    """Returns a list of problem strings (empty = healthy).  CPU-only
    environments trivially pass."""
    problems: List[str] = []
    if not torch.cuda.is_available():
        return problems
    devices = [device] if device is not None else [
        torch.device(f"cuda:{i}") for i in range(torch.cuda.device_count())
    ]
    torch.manual_seed(1234)
    a = torch.randn(size, size)
    b = torch.randn(size, size)
    ref = (a @ b).float()
    ref_norm = ref.norm()
    for dev in devices:
        try:
            ag = a.to(dev, torch.bfloat16)
            bg = b.to(dev, torch.bfloat16)
            cg = (ag @ bg).float().cpu()
            rel = (cg - ref).norm() / ref_norm
            if not torch.isfinite(cg).all():
                problems.append(f"{dev}: non-finite GEMM output")
            elif float(rel) > tol:
                problems.append(f"{dev}: GEMM relative error {float(rel):.3e} > {tol}")
            # repeatability: a second run must match the first bit-for-bit
            cg2 = (ag @ bg).float().cpu()
            if not torch.equal(cg, cg2):
                problems.append(f"{dev}: non-deterministic GEMM (flaky hardware?)")
        except RuntimeError as e:
            problems.append(f"{dev}: {e}")
    return problems
