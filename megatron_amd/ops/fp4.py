"""MXFP4 (OCP microscaling fp4) quantization utilities.

Capability analog of reference megatron/core/fp4_utils.py (nvfp4 recipe).
CDNA4's MFMA has native MXFP4/MXFP6 support (gfx950
V_MFMA_*_F8F6F4 with per-32-element E8M0 shared scales); these helpers
implement the OCP MX quantization rule so weights/activations can be
prepared for those instructions and so CPU tests can check the numerics:

  * element format E2M1: values {0, .5, 1, 1.5, 2, 3, 4, 6} x sign
  * block size 32 along the last dim, one shared power-of-two scale per
    block (E8M0), chosen so the block max maps to the largest code (6).
"""

from __future__ import annotations

from typing import Tuple

import torch

# the 8 non-negative E2M1 magnitudes
FP4_VALUES = torch.tensor([0.0, 0.5, 1.0, 1.5, 2.0, 3.0, 4.0, 6.0])
FP4_MAX = 6.0
MX_BLOCK = 32


def _round_to_fp4_grid(x: torch.Tensor) -> torch.Tensor:
    """Round |x| (already scaled into [0, 6]) to the nearest E2M1 magnitude,
    ties-to-even on the code index like hardware RNE."""
    grid = FP4_VALUES.to(x.device, x.dtype)
    # midpoints between consecutive grid values
    mid = (grid[1:] + grid[:-1]) / 2
    idx = torch.bucketize(x.abs(), mid)
    return torch.sign(x) * grid[idx]


def quantize_mxfp4(t: torch.Tensor, block: int = MX_BLOCK) -> Tuple[torch.Tensor, torch.Tensor]:
    """Quantize along the last dim.  Returns (q, scales):
    q   — same shape as t, fp32 holding exact E2M1 values (pre-scale),
    scales — [..., n_blocks] power-of-two E8M0 scales.
    dequantize = q * scales (broadcast per block)."""
    orig_shape = t.shape
    n = orig_shape[-1]
    assert n % block == 0, (n, block)
    x = t.float().reshape(*orig_shape[:-1], n // block, block)
    amax = x.abs().amax(dim=-1, keepdim=True).clamp(min=1e-30)
    # E8M0: scale is a pure power of two; pick 2^ceil(log2(amax/6))
    exp = torch.ceil(torch.log2(amax / FP4_MAX))
    scale = torch.exp2(exp)
    q = _round_to_fp4_grid(x / scale)
    return q.reshape(orig_shape), scale.squeeze(-1)


def dequantize_mxfp4(q: torch.Tensor, scales: torch.Tensor, block: int = MX_BLOCK) -> torch.Tensor:
    orig_shape = q.shape
    n = orig_shape[-1]
    x = q.reshape(*orig_shape[:-1], n // block, block)
    return (x * scales.unsqueeze(-1)).reshape(orig_shape)


def pack_fp4_codes(q: torch.Tensor) -> torch.Tensor:
    """Encode E2M1 values to 4-bit codes packed two-per-byte (uint8), the
    in-memory layout the MFMA F8F6F4 path consumes."""
    grid = FP4_VALUES.to(q.device)
    flat = q.reshape(-1)
    mag_idx = (flat.abs().unsqueeze(-1) - grid).abs().argmin(dim=-1)  # [N] in 0..7
    sign = (flat < 0).to(torch.uint8)
    codes = (sign << 3) | mag_idx.to(torch.uint8)
    assert codes.numel() % 2 == 0
    lo = codes[0::2]
    hi = codes[1::2]
    return (hi << 4) | lo


def unpack_fp4_codes(packed: torch.Tensor, numel: int) -> torch.Tensor:
    grid = FP4_VALUES.to(packed.device)
    lo = packed & 0xF
    hi = packed >> 4
    codes = torch.stack([lo, hi], dim=-1).reshape(-1)[:numel]
    mag = grid[(codes & 0x7).long()]
    sign = torch.where((codes >> 3) > 0, -1.0, 1.0)
    return sign * mag


def mxfp4_quantization_error(t: torch.Tensor) -> float:
    """Relative RMS error of an mxfp4 round trip (diagnostics)."""
    q, s = quantize_mxfp4(t)
    back = dequantize_mxfp4(q, s)
    return float((back - t.float()).pow(2).mean().sqrt() / t.float().pow(2).mean().sqrt().clamp(min=1e-30))
