"""Per-process-group RCCL communicator tuning.

Capability analog of reference parallel_state.py:168 `get_nccl_options` +
process_groups_config.py (`--nccl-communicator-config-path` yaml): each
named group (tp/dp/pp/ep/...) can pin RCCL CTA counts and stream priority —
on MI355X the interesting knobs are max_ctas (CUs ceded to communication
kernels: fewer CTAs leave more CUs for overlapped MFMA work) and
high-priority streams for the latency-critical TP collectives.

yaml shape (same as the reference's):

    tp: {min_ctas: 4, max_ctas: 8, is_high_priority_stream: true}
    dp: {max_ctas: 16}

Usage: `load_comm_config(path)` once, then `pg_options_for("tp")` wherever
a group is created (grid.py consults it when a config is loaded).  Returns
None when unsupported (gloo, CPU) — callers pass it straight to
`dist.new_group(pg_options=...)`.
"""

from __future__ import annotations

from typing import Dict, Optional

import torch
import torch.distributed as dist

_CONFIG: Dict[str, dict] = {}


def load_comm_config(path_or_dict) -> Dict[str, dict]:
    global _CONFIG
    if isinstance(path_or_dict, dict):
        _CONFIG = dict(path_or_dict)
    else:
        import yaml

        with open(path_or_dict) as f:
            _CONFIG = yaml.safe_load(f) or {}
    for name, knobs in _CONFIG.items():
        unknown = set(knobs) - {"min_ctas", "max_ctas", "cga_cluster_size",
                                "is_high_priority_stream"}
        if unknown:
            raise ValueError(f"unknown comm-config knobs for {name!r}: {unknown}")
    return _CONFIG


def comm_config() -> Dict[str, dict]:
    return _CONFIG


def pg_options_for(group_name: str):
    """ProcessGroupNCCL.Options for a named group, or None."""
    knobs = _CONFIG.get(group_name)
    if not knobs:
        return None
    if not torch.cuda.is_available():
        return None  # gloo path ignores options
    try:
        from torch.distributed import ProcessGroupNCCL
    except ImportError:
        return None
    opts = ProcessGroupNCCL.Options()
    opts.is_high_priority_stream = bool(knobs.get("is_high_priority_stream", False))
    if "min_ctas" in knobs:
        opts.config.min_ctas = int(knobs["min_ctas"])
    if "max_ctas" in knobs:
        opts.config.max_ctas = int(knobs["max_ctas"])
    if "cga_cluster_size" in knobs:
        opts.config.cga_cluster_size = int(knobs["cga_cluster_size"])
    return opts
