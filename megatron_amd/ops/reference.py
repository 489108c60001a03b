"""Plain-PyTorch reference implementations of every fused HIP op.

These are the numerics oracles (tests compare the HIP kernels against these
in fp32) and the CPU execution path.  They are NOT used on GPU unless the
native extension is deliberately disabled — on a GPU box a missing extension
raises (see ops/__init__.py).
"""

from __future__ import annotations

import math
from typing import Optional, Tuple

import torch
import torch.nn.functional as F


def rms_norm(x: torch.Tensor, weight: torch.Tensor, eps: float) -> torch.Tensor:
    dtype = x.dtype
    xf = x.float()
    var = xf.pow(2).mean(dim=-1, keepdim=True)
    out = xf * torch.rsqrt(var + eps)
    return (out * weight.float()).to(dtype)


def layer_norm(x: torch.Tensor, weight: torch.Tensor, bias: Optional[torch.Tensor], eps: float) -> torch.Tensor:
    dtype = x.dtype
    out = F.layer_norm(x.float(), (x.size(-1),), weight.float(), bias.float() if bias is not None else None, eps)
    return out.to(dtype)


def swiglu(x: torch.Tensor) -> torch.Tensor:
    """x: [..., 2*ffn]; llama convention: silu(x1) * x2 with x1 = first half."""
    x1, x2 = x.chunk(2, dim=-1)
    return (F.silu(x1.float()) * x2.float()).to(x.dtype)


def geglu(x: torch.Tensor) -> torch.Tensor:
    x1, x2 = x.chunk(2, dim=-1)
    return (F.gelu(x1.float(), approximate="tanh") * x2.float()).to(x.dtype)


def squared_relu(x: torch.Tensor) -> torch.Tensor:
    return torch.pow(F.relu(x.float()), 2).to(x.dtype)


def apply_llama3_rope_scaling(inv_freq: torch.Tensor, factor: float = 8.0,
                              low_freq_factor: float = 1.0,
                              high_freq_factor: float = 4.0,
                              original_max_position: int = 8192) -> torch.Tensor:
    """Llama-3.1 long-context rope scaling (reference rope_utils /
    HF _compute_llama3_parameters): high-frequency channels untouched,
    low-frequency channels divided by `factor`, smooth ramp between."""
    low_wavelen = original_max_position / low_freq_factor
    high_wavelen = original_max_position / high_freq_factor
    wavelen = 2 * math.pi / inv_freq
    scaled = torch.where(wavelen > low_wavelen, inv_freq / factor, inv_freq)
    smooth = (original_max_position / wavelen - low_freq_factor) / (
        high_freq_factor - low_freq_factor)
    smoothed = (1 - smooth) / factor * inv_freq + smooth * inv_freq
    mid = (wavelen <= low_wavelen) & (wavelen >= high_wavelen)
    return torch.where(mid, smoothed, scaled)


def rope_freqs(
    seq_len: int,
    dim: int,
    base: float = 10000.0,
    device=None,
    dtype=torch.float32,
    rotary_percent: float = 1.0,
    rope_scaling: dict = None,
) -> torch.Tensor:
    """Precomputed rotation angles [seq, dim_rot/2] (host-side table — G13/App-B:
    on-device trig turns RoPE memory-bound into VALU-bound).  rope_scaling:
    None or {'type': 'llama3', 'factor': ..., ...} (llama-3.1 long context)."""
    rot_dim = int(dim * rotary_percent)
    inv_freq = 1.0 / (base ** (torch.arange(0, rot_dim, 2, device=device, dtype=torch.float32) / rot_dim))
    if rope_scaling:
        kind = rope_scaling.get("type", "llama3")
        if kind == "llama3":
            inv_freq = apply_llama3_rope_scaling(
                inv_freq,
                factor=rope_scaling.get("factor", 8.0),
                low_freq_factor=rope_scaling.get("low_freq_factor", 1.0),
                high_freq_factor=rope_scaling.get("high_freq_factor", 4.0),
                original_max_position=rope_scaling.get(
                    "original_max_position_embeddings", 8192),
            )
        elif kind == "linear":
            inv_freq = inv_freq / rope_scaling.get("factor", 1.0)
        elif kind == "yarn":
            inv_freq = apply_yarn_rope_scaling(
                inv_freq,
                factor=rope_scaling.get("factor", 8.0),
                beta_fast=rope_scaling.get("beta_fast", 32.0),
                beta_slow=rope_scaling.get("beta_slow", 1.0),
                original_max_position=rope_scaling.get(
                    "original_max_position_embeddings", 4096),
                base=base, rot_dim=rot_dim,
            )
        else:
            raise ValueError(f"unknown rope_scaling type {kind!r}")
    t = torch.arange(seq_len, device=device, dtype=torch.float32)
    freqs = torch.outer(t, inv_freq)  # [seq, rot_dim/2]
    return freqs.to(dtype)


def yarn_mscale(factor: float) -> float:
    """YaRN attention-temperature: each of q/k cos/sin is scaled by
    0.1*ln(factor)+1, i.e. scores by its square."""
    import math as _m

    return 0.1 * _m.log(factor) + 1.0 if factor > 1.0 else 1.0


def apply_yarn_rope_scaling(inv_freq, factor: float, beta_fast: float,
                            beta_slow: float, original_max_position: int,
                            base: float, rot_dim: int):
    """NTK-by-parts interpolation (YaRN): high-frequency dims (short
    wavelengths, fully inside the original window beta_fast times) keep
    their frequency; low-frequency dims interpolate by 1/factor; the band
    between ramps linearly in dimension index."""
    import math as _m

    def find_dim(num_rotations):
        # dim index whose wavelength fits `num_rotations` times in the window
        return (rot_dim * _m.log(original_max_position / (num_rotations * 2 * _m.pi))
                / (2 * _m.log(base)))

    lo = max(_m.floor(find_dim(beta_fast)), 0)
    hi = min(_m.ceil(find_dim(beta_slow)), rot_dim // 2 - 1)
    ramp = ((torch.arange(rot_dim // 2, dtype=torch.float32, device=inv_freq.device) - lo)
            / max(hi - lo, 1)).clamp(0.0, 1.0)
    keep = 1.0 - ramp  # 1 for high-freq dims, 0 for low-freq
    return inv_freq * keep + (inv_freq / factor) * ramp


def mrope_freqs(position_ids: torch.Tensor, dim: int, base: float = 10000.0,
                mrope_section=(16, 24, 24), rotary_percent: float = 1.0) -> torch.Tensor:
    """Multimodal rotary freqs (Qwen2-VL style mrope; reference
    models/common/embeddings multimodal rotary :266 analog).

    position_ids: [3, s] — (temporal, height, width) position per token
    (text tokens carry the same value in all three rows).  The rotation
    channels are split into `mrope_section` groups; group j's angles are
    indexed by position row j.  Returns [s, dim_rot/2] angles usable by
    rope_apply."""
    assert position_ids.dim() == 2 and position_ids.shape[0] == len(mrope_section)
    rot_dim = int(dim * rotary_percent)
    n_half = rot_dim // 2
    assert sum(mrope_section) == n_half, (mrope_section, n_half)
    inv_freq = 1.0 / (base ** (torch.arange(0, rot_dim, 2, device=position_ids.device,
                                            dtype=torch.float32) / rot_dim))
    s = position_ids.shape[1]
    out = torch.empty(s, n_half, dtype=torch.float32, device=position_ids.device)
    start = 0
    for row, width in enumerate(mrope_section):
        t = position_ids[row].float()
        out[:, start:start + width] = torch.outer(t, inv_freq[start:start + width])
        start += width
    return out


def rope_apply(x: torch.Tensor, freqs: torch.Tensor) -> torch.Tensor:
    """Rotate-half RoPE (GPT-NeoX / llama convention).

    x: [s, b, h, d]; freqs: [s, d_rot/2] (fp32).  First d_rot channels rotated.
    """
    s, b, h, d = x.shape
    d_rot = freqs.size(1) * 2
    xf = x.float()
    x_rot, x_pass = xf[..., :d_rot], xf[..., d_rot:]
    cos = torch.cos(freqs).view(s, 1, 1, -1)
    sin = torch.sin(freqs).view(s, 1, 1, -1)
    x1, x2 = x_rot[..., : d_rot // 2], x_rot[..., d_rot // 2 :]
    out_rot = torch.cat((x1 * cos - x2 * sin, x2 * cos + x1 * sin), dim=-1)
    if d_rot < d:
        out_rot = torch.cat((out_rot, x_pass), dim=-1)
    return out_rot.to(x.dtype)


def rope_apply_per_row(x: torch.Tensor, freqs: torch.Tensor) -> torch.Tensor:
    """Rotate-half RoPE with per-batch-row angles.

    x: [s, b, h, d]; freqs: [s, b, d_rot/2] (fp32) — each sequence in the
    batch sits at its own position (continuous-batching decode)."""
    s, b, h, d = x.shape
    d_rot = freqs.size(-1) * 2
    xf = x.float()
    x_rot, x_pass = xf[..., :d_rot], xf[..., d_rot:]
    cos = torch.cos(freqs).view(s, b, 1, -1)
    sin = torch.sin(freqs).view(s, b, 1, -1)
    x1, x2 = x_rot[..., : d_rot // 2], x_rot[..., d_rot // 2 :]
    out_rot = torch.cat((x1 * cos - x2 * sin, x2 * cos + x1 * sin), dim=-1)
    if d_rot < d:
        out_rot = torch.cat((out_rot, x_pass), dim=-1)
    return out_rot.to(x.dtype)


def bias_dropout_add(
    x: torch.Tensor, bias: Optional[torch.Tensor], residual: torch.Tensor, p: float, training: bool
) -> torch.Tensor:
    if bias is not None:
        x = x + bias
    if p > 0.0:
        x = F.dropout(x, p=p, training=training)
    return residual + x


def attention_varlen(
    q: torch.Tensor,
    k: torch.Tensor,
    v: torch.Tensor,
    cu_seqlens: torch.Tensor,
    causal: bool = True,
    scale: Optional[float] = None,
) -> torch.Tensor:
    """Packed (THD) attention: q/k/v [t, 1, h, d] with document boundaries
    at cu_seqlens; each token attends only within its own document
    (block-diagonal causal mask).  fp32 softmax, GQA via repeat."""
    t, b, hq, d = q.shape
    assert b == 1, "packed sequences use batch 1 (thd layout)"
    hkv = k.shape[2]
    rep = hq // hkv
    if scale is None:
        scale = 1.0 / math.sqrt(d)
    qf = q.permute(1, 2, 0, 3).float()  # [1, hq, t, d]
    kf = k.permute(1, 2, 0, 3).float()
    vf = v.permute(1, 2, 0, 3).float()
    if rep > 1:
        kf = kf.repeat_interleave(rep, dim=1)
        vf = vf.repeat_interleave(rep, dim=1)
    scores = torch.matmul(qf, kf.transpose(-1, -2)) * scale  # [1, hq, t, t]
    seg = torch.bucketize(torch.arange(t, device=q.device), cu_seqlens[1:-1], right=True)
    mask = seg.view(t, 1) == seg.view(1, t)
    if causal:
        mask = mask & torch.ones(t, t, dtype=torch.bool, device=q.device).tril_()
    scores = scores.masked_fill(~mask, float("-inf"))
    probs = torch.softmax(scores, dim=-1)
    out = torch.matmul(probs, vf)  # [1, hq, t, d]
    return out.permute(2, 0, 1, 3).to(q.dtype)


def attention(
    q: torch.Tensor,
    k: torch.Tensor,
    v: torch.Tensor,
    causal: bool = True,
    scale: Optional[float] = None,
    window: Optional[int] = None,
    dropout_p: float = 0.0,
    training: bool = False,
    bias: Optional[torch.Tensor] = None,
) -> torch.Tensor:
    """Unfused reference attention.

    q: [s, b, hq, d], k/v: [s, b, hkv, d] -> out [s, b, hq, d].
    GQA: hq is a multiple of hkv.  fp32 softmax; optional attention-prob
    dropout (the reference's attention_dropout — applied post-softmax)."""
    s, b, hq, d = q.shape
    hkv = k.shape[2]
    rep = hq // hkv
    if scale is None:
        scale = 1.0 / math.sqrt(d)
    qf = q.permute(1, 2, 0, 3).float()  # [b, hq, s, d]
    kf = k.permute(1, 2, 0, 3).float()
    vf = v.permute(1, 2, 0, 3).float()
    if rep > 1:
        kf = kf.repeat_interleave(rep, dim=1)
        vf = vf.repeat_interleave(rep, dim=1)
    scores = torch.matmul(qf, kf.transpose(-1, -2)) * scale  # [b, hq, s, s]
    if bias is not None:
        scores = scores + bias.float()  # additive [hq, s, sk] (T5 relative bias)
    sk = k.shape[0]
    off = sk - s  # decode: query row i attends kv <= i + off
    if causal:
        mask = torch.ones(s, sk, dtype=torch.bool, device=q.device).tril_(off)
        if window is not None:
            mask &= torch.ones(s, sk, dtype=torch.bool, device=q.device).triu_(off - window + 1)
        scores = scores.masked_fill(~mask, float("-inf"))
    probs = torch.softmax(scores, dim=-1)
    if dropout_p > 0.0 and training:
        probs = torch.nn.functional.dropout(probs, p=dropout_p)
    out = torch.matmul(probs, vf)  # [b, hq, s, d]
    return out.permute(2, 0, 1, 3).to(q.dtype)


def attention_padded(q, k, v, key_valid_mask, causal: bool = False, scale=None,
                     bias=None):
    """Attention with a [b, s_k] key-padding mask (True = valid token).

    Capability analog of the reference's arbitrary-mask (non-flash) path
    (fused_softmax scaled_masked variant): used by BERT-style bidirectional
    batches with right padding.  Optional additive `bias` [hq, s_q, s_k]
    (T5 relative bias on padded batches).  fp32 softmax; [s,b,h,d]."""
    import math as _math

    if scale is None:
        scale = 1.0 / _math.sqrt(q.shape[-1])
    s_q, b, hq, d = q.shape
    hkv = k.shape[2]
    rep = hq // hkv
    qf = q.permute(1, 2, 0, 3).float()
    kf = k.permute(1, 2, 0, 3).float().repeat_interleave(rep, dim=1)
    vf = v.permute(1, 2, 0, 3).float().repeat_interleave(rep, dim=1)
    scores = torch.matmul(qf, kf.transpose(-1, -2)) * scale  # [b, hq, s_q, s_k]
    if bias is not None:
        scores = scores + bias.float()
    bad = ~key_valid_mask.to(torch.bool)                     # [b, s_k]
    scores = scores.masked_fill(bad[:, None, None, :], float("-inf"))
    if causal:
        s_k = k.shape[0]
        cm = torch.ones(s_q, s_k, dtype=torch.bool, device=q.device).tril_(s_k - s_q)
        scores = scores.masked_fill(~cm, float("-inf"))
    probs = torch.softmax(scores, dim=-1)
    probs = torch.nan_to_num(probs)  # fully-masked rows (pad queries) -> 0
    out = torch.matmul(probs, vf)
    return out.permute(2, 0, 1, 3).to(q.dtype)
