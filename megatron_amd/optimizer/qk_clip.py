"""QK-clip (MuonClip-style attention-logit growth control).

Capability analog of reference megatron/core/optimizer/qk_clip.py: after an
optimizer step, any attention head whose max pre-softmax logit exceeded a
threshold tau gets its Q and K projection rows scaled by
sqrt(tau / max_logit), pulling the logits back under tau without touching
the attention output distribution shape.  Used with Muon, whose
orthogonalized updates are known to inflate attention logits.

Mechanics here: SelfAttention records per-query-group max logits during
forward when `config.qk_clip_threshold` is set (an extra amax-only QK^T
pass under no_grad — qk-clip is opt-in).  `apply_qk_clip` then reduces the
stat over DP ranks (MAX) and scales the q/k rows of the fused QKV weight
per query group.
"""

from __future__ import annotations

import math
from typing import Dict, List

import torch
import torch.distributed as dist

from megatron_amd.parallel import grid as G


@torch.no_grad()
def max_logits_per_group(q: torch.Tensor, k: torch.Tensor, scale: float) -> torch.Tensor:
    """q [s,b,hq,d], k [s,b,hkv,d] -> max |logit| per kv group [hkv]
    (max over batch, positions and the group's query heads)."""
    s, b, hq, d = q.shape
    hkv = k.shape[2]
    rep = hq // hkv
    qg = q.view(s, b, hkv, rep, d).permute(1, 2, 3, 0, 4)   # [b,hkv,rep,s,d]
    kg = k.permute(1, 2, 0, 3)                               # [b,hkv,s,d]
    logits = torch.einsum("bgrsd,bgtd->bgrst", qg.float(), kg.float()) * scale
    return logits.abs().amax(dim=(0, 2, 3, 4))               # [hkv]


@torch.no_grad()
def apply_qk_clip(model: torch.nn.Module, threshold: float) -> int:
    """Scale q/k weight rows of every attention whose recorded max logit
    exceeded `threshold`.  Returns the number of clipped groups."""
    from megatron_amd.transformer.attention import SelfAttention

    clipped = 0
    for m in model.modules():
        if not isinstance(m, SelfAttention):
            continue
        stats = getattr(m, "last_max_logit", None)
        if stats is None:
            continue
        if dist.is_initialized() and G.grid_initialized():
            group = G.get_grid().group("dp_cp")
            if group is not None and dist.get_world_size(group) > 1:
                dist.all_reduce(stats, op=dist.ReduceOp.MAX, group=group)
        gamma = (threshold / stats.clamp(min=1e-20)).clamp(max=1.0)  # [ng]
        if bool((gamma >= 1.0).all()):
            continue
        root = math.sqrt  # scale q and k each by sqrt(gamma)
        ng = m.num_query_groups_per_partition
        rep = m.num_heads_per_partition // ng
        d = m.kv_channels
        w = m.linear_qkv.weight  # [(rep+2)*d*ng, h] rows grouped per kv group
        per_group = (rep + 2) * d
        for g in range(ng):
            gval = float(gamma[g])
            if gval >= 1.0:
                continue
            base = g * per_group
            sq = root(gval)
            w[base: base + rep * d].mul_(sq)            # q rows of this group
            w[base + rep * d: base + (rep + 1) * d].mul_(sq)  # k rows
            clipped += 1
        m.last_max_logit = None
    return clipped
