"""Per-module activation / gradient debug dumps.

Capability analog of reference activation_logging.py / dgrad_logging.py
(+ save_wgrads/params intervals in train_step): forward/backward hooks on
named modules record per-tensor statistics (norm, absmax, mean, NaN/Inf
counts) to a JSONL stream, for chasing numerics divergence between runs or
ranks without a debugger attached to 8 GPUs."""

from __future__ import annotations

import json
from typing import Dict, List, Optional

import torch


def _stats(t: torch.Tensor) -> dict:
    tf = t.detach().float()
    finite = torch.isfinite(tf)
    return {
        "shape": list(t.shape),
        "norm": float(tf[finite].norm()) if finite.any() else 0.0,
        "absmax": float(tf[finite].abs().max()) if finite.any() else 0.0,
        "mean": float(tf[finite].mean()) if finite.any() else 0.0,
        "n_nonfinite": int((~finite).sum()),
    }


class DebugDumper:
    """Attach with `watch(model, patterns)`; every matched module logs its
    forward output (and, with `grads=True`, its grad_output) each step."""

    def __init__(self, path: Optional[str] = None, rank: int = 0):
        self.path = path
        self.rank = rank
        self.records: List[dict] = []  # kept in memory too (tests, quick looks)
        self._handles = []
        self._fh = open(path, "a") if path else None
        self.step = 0

    def _emit(self, kind: str, name: str, tensor: torch.Tensor):
        rec = {"step": self.step, "rank": self.rank, "kind": kind, "module": name,
               **_stats(tensor)}
        self.records.append(rec)
        if self._fh is not None:
            self._fh.write(json.dumps(rec) + "\n")
            self._fh.flush()

    def watch(self, model: torch.nn.Module, patterns: List[str], grads: bool = False):
        import fnmatch

        n = 0
        for name, mod in model.named_modules():
            if not any(fnmatch.fnmatch(name, p) for p in patterns):
                continue
            n += 1

            def fwd_hook(m, inp, out, _name=name):
                t = out[0] if isinstance(out, (tuple, list)) else out
                if torch.is_tensor(t):
                    self._emit("activation", _name, t)

            self._handles.append(mod.register_forward_hook(fwd_hook))
            if grads:
                def bwd_hook(m, gin, gout, _name=name):
                    t = gout[0] if isinstance(gout, (tuple, list)) else gout
                    if torch.is_tensor(t):
                        self._emit("dgrad", _name, t)

                self._handles.append(mod.register_full_backward_hook(bwd_hook))
        return n

    def next_step(self):
        self.step += 1

    def nonfinite_modules(self) -> List[str]:
        return sorted({r["module"] for r in self.records if r["n_nonfinite"] > 0})

    def close(self):
        for h in self._handles:
            h.remove()
        self._handles = []
        if self._fh is not None:
            self._fh.close()
            self._fh = None
