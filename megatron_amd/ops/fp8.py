"""FP8 training path for CDNA4 (K15 in SURVEY.md §2.3).

Capability analog of reference core/fp8_utils.py + TE recipes: per-tensor
delayed scaling (amax history -> scale), hybrid formats (e4m3 for activations
and weights, e5m2 for gradients — OCP encodings, the gfx950 native ones), and
an fp8 linear autograd function whose forward and dgrad GEMMs run through
hipBLASLt's fp8 path (torch._scaled_mm); the wgrad keeps the bf16
fp32-accumulate path (K9) for accuracy.

MI355X note: non-block-scaled fp8 MFMA runs at the bf16 rate; the fp8 win
here is halved operand traffic (HBM/LDS/L2) on the GEMM inputs.
"""

from __future__ import annotations

from typing import Optional

import torch

E4M3_MAX = 448.0
E5M2_MAX = 57344.0


class DelayedScaling:
    """Per-tensor amax history -> scale (reference TE DelayedScaling recipe).

    scale = fp8_max / (2^margin * max(amax_history)); history updated every
    step with the current amax."""

    def __init__(self, history_len: int = 16, margin: int = 0, fmt_max: float = E4M3_MAX):
        self.history_len = history_len
        self.margin = margin
        self.fmt_max = fmt_max
        self._history: Optional[torch.Tensor] = None
        self._pos = 0

    def scale_for(self, t: torch.Tensor) -> torch.Tensor:
        amax = t.detach().abs().max().float().clamp(min=1e-12)
        # DP-consistent scales (reference fp8 amax group reduction,
        # parallel_state.py:1877): replicas must quantize identically or
        # their updates diverge
        import torch.distributed as dist

        if dist.is_initialized() and dist.get_world_size() > 1:
            from megatron_amd.parallel import grid as G

            if G.grid_initialized() and G.get_data_parallel_world_size(with_context_parallel=True) > 1:
                dist.all_reduce(amax, op=dist.ReduceOp.MAX,
                                group=G.get_grid().group("dp_cp"))
        if self._history is None:
            self._history = torch.zeros(self.history_len, device=t.device)
        if self._history.device != t.device:
            self._history = self._history.to(t.device)
        # use the running max BEFORE inserting the current amax (delayed);
        # first call falls back to the current amax
        hist_max = self._history.max()
        eff = torch.where(hist_max > 0, hist_max, amax)
        self._history[self._pos % self.history_len] = amax
        self._pos += 1
        return self.fmt_max / (eff * (2.0 ** self.margin))


def quantize_fp8(t: torch.Tensor, scale: torch.Tensor, fmt=torch.float8_e4m3fn):
    """returns (fp8 tensor, inv_scale fp32 scalar) with saturation clamp."""
    fmax = E4M3_MAX if fmt == torch.float8_e4m3fn else E5M2_MAX
    q = (t.float() * scale).clamp(-fmax, fmax).to(fmt)
    return q, scale.reciprocal()


def _scaled_mm(a_fp8, b_fp8, inv_a, inv_b, out_dtype):
    """a [m,k] row-major fp8, b [k,n] fp8 (made column-major) -> [m,n]."""
    b_cm = b_fp8.t().contiguous().t()  # hipBLASLt wants mat2 column-major
    return torch._scaled_mm(a_fp8, b_cm, scale_a=inv_a, scale_b=inv_b, out_dtype=out_dtype)


class _Fp8LinearFn(torch.autograd.Function):
    """y = x @ w^T with fp8 forward and dgrad GEMMs.

    x: [*, in], w: [out, in]. Forward: e4m3 x @ e4m3 w^T. dgrad: e5m2 dy @
    e4m3 w. wgrad: bf16 fp32-accum (main_grad path handled by the caller).
    """

    @staticmethod
    def forward(ctx, x, w, recipes):
        rx, rw, rg = recipes
        x2d = x.reshape(-1, x.shape[-1])
        sx = rx.scale_for(x2d)
        sw = rw.scale_for(w)
        xq, inv_x = quantize_fp8(x2d, sx)
        wq, inv_w = quantize_fp8(w, sw)
        out = _scaled_mm(xq, wq.t(), inv_x, inv_w, x.dtype)
        ctx.save_for_backward(x2d, w, wq, inv_w)
        ctx.rg = rg
        ctx.in_shape = x.shape
        return out.reshape(*x.shape[:-1], w.shape[0])

    @staticmethod
    def backward(ctx, dy):
        x2d, w, wq, inv_w = ctx.saved_tensors
        dy2d = dy.reshape(-1, dy.shape[-1])
        sg = ctx.rg.scale_for(dy2d)
        dyq, inv_dy = quantize_fp8(dy2d, sg, fmt=torch.float8_e5m2)
        dx = _scaled_mm(dyq, wq, inv_dy, inv_w, dy.dtype)  # [tokens, in]
        dw = torch.matmul(dy2d.t(), x2d)  # bf16 wgrad (accuracy-critical)
        return dx.reshape(ctx.in_shape), dw, None


def fp8_linear(x: torch.Tensor, w: torch.Tensor, recipes) -> torch.Tensor:
    return _Fp8LinearFn.apply(x, w, recipes)


def fp8_eligible(x: torch.Tensor, w: torch.Tensor) -> bool:
    """hipBLASLt fp8 GEMM wants 16-aligned shapes."""
    tokens = x.numel() // x.shape[-1]
    return (x.is_cuda and x.shape[-1] % 16 == 0 and w.shape[0] % 16 == 0
            and tokens % 16 == 0)


def make_recipes(fmt: str = "hybrid", history_len: int = 16, margin: int = 0):
    """(input, weight, grad) recipes. 'hybrid' = e4m3 fwd + e5m2 grads
    (reference TE Format.HYBRID); 'e4m3' clamps grads to e4m3 range too."""
    gmax = E5M2_MAX if fmt == "hybrid" else E4M3_MAX
    return (DelayedScaling(history_len, margin, E4M3_MAX),
            DelayedScaling(history_len, margin, E4M3_MAX),
            DelayedScaling(history_len, margin, gmax))
