"""Mamba2 mixer (selective SSM with SSD scan).

Capability analog of reference megatron/core/ssm/mamba_mixer.py:144
(MambaMixer): fused in_proj -> [z | x | B | C | dt] split, depthwise causal
conv over [x B C], softplus dt, chunked SSD scan, gated group-RMSNorm,
out_proj.  TP shards heads and B/C groups column-wise in in_proj/conv and
row-wise in out_proj, exactly mirroring the reference's sharding contract
(nheads % tp == 0, ngroups % tp == 0) so no extra collectives are needed
inside the mixer.

MI355X notes: the SSD scan (ssm/ssd.py) is all batched GEMMs -> MFMA via
rocBLAS; the conv is a depthwise F.conv1d (MIOpen); state math is fp32.
Decode keeps a rolling conv window and an fp32 SSM state per request
(reference's inference_params conv_state/ssm_state contract).
"""

from __future__ import annotations

import math
from typing import Optional, Tuple

import torch
import torch.nn as nn
import torch.nn.functional as F

from megatron_amd.parallel import grid as G
from megatron_amd.parallel.layers import ColumnParallelLinear, RowParallelLinear
from megatron_amd.ssm.ssd import ssd_chunked_scan, ssd_step


class GatedRMSNorm(nn.Module):
    """RMSNorm(y * silu(z)) computed per B/C group (reference RMSNormGated,
    mamba_mixer.py:114 ExtendedRMSNorm).  Groups are TP-local so the
    normalization needs no cross-rank reduction."""

    def __init__(self, d_local: int, ngroups_local: int, eps: float, dtype: torch.dtype):
        super().__init__()
        self.eps = eps
        self.ngroups_local = ngroups_local
        self.weight = nn.Parameter(torch.ones(d_local, dtype=dtype))
        self.weight.tensor_parallel = True
        self.weight.partition_dim = 0

    def forward(self, y: torch.Tensor, z: torch.Tensor) -> torch.Tensor:
        dt = y.dtype
        y = y.float() * F.silu(z.float())
        g = y.view(*y.shape[:-1], self.ngroups_local, -1)
        g = g * torch.rsqrt(g.pow(2).mean(-1, keepdim=True) + self.eps)
        return (g.reshape_as(y) * self.weight.float()).to(dt)


class MambaMixer(nn.Module):
    def __init__(self, config, layer_number: int = 0):
        super().__init__()
        self.config = config
        self.layer_number = layer_number
        tp = G.get_tensor_model_parallel_world_size()

        self.d_model = config.hidden_size
        self.d_state = config.mamba_state_dim
        self.d_conv = config.mamba_d_conv
        self.headdim = config.mamba_head_dim
        self.d_inner = config.mamba_expand * self.d_model
        assert self.d_inner % self.headdim == 0
        self.nheads = self.d_inner // self.headdim
        self.ngroups = config.mamba_num_groups
        assert self.nheads % self.ngroups == 0
        assert self.nheads % tp == 0 and self.ngroups % tp == 0, (
            "Mamba TP shards heads and groups: need nheads % tp == 0 and ngroups % tp == 0"
        )
        self.chunk_size = config.mamba_chunk_size

        self.d_inner_local = self.d_inner // tp
        self.nheads_local = self.nheads // tp
        self.ngroups_local = self.ngroups // tp
        self.conv_dim_local = self.d_inner_local + 2 * self.ngroups_local * self.d_state

        # Fused projection; each rank's shard is laid out [z|x|B|C|dt] locally
        # (we own weight creation, so row semantics are ours to define — the
        # sharded checkpoint stores this as a TP-concatenated atlas like the
        # fused QKV / gated-fc1 weights).
        d_proj = 2 * self.d_inner + 2 * self.ngroups * self.d_state + self.nheads
        self.in_proj = ColumnParallelLinear(
            self.d_model, d_proj, config=config, bias=config.add_linear_bias,
            gather_output=False, skip_bias_add=False,
        )
        self.split_sizes = [
            self.d_inner_local,                      # z
            self.d_inner_local,                      # x
            self.ngroups_local * self.d_state,       # B
            self.ngroups_local * self.d_state,       # C
            self.nheads_local,                       # dt
        ]

        pdtype = config.params_dtype
        self.conv_weight = nn.Parameter(torch.empty(self.conv_dim_local, 1, self.d_conv, dtype=pdtype))
        self.conv_bias = nn.Parameter(torch.zeros(self.conv_dim_local, dtype=pdtype))
        for p in (self.conv_weight, self.conv_bias):
            p.tensor_parallel = True
            p.partition_dim = 0
        with torch.no_grad():
            self.conv_weight.uniform_(-math.sqrt(1 / self.d_conv), math.sqrt(1 / self.d_conv))

        # dt bias: softplus(dt_bias) uniform in [dt_min, dt_max] (reference :375)
        dt_min, dt_max, dt_floor = 1e-3, 0.1, 1e-4
        dt = torch.exp(
            torch.rand(self.nheads_local) * (math.log(dt_max) - math.log(dt_min)) + math.log(dt_min)
        ).clamp(min=dt_floor)
        inv_dt = dt + torch.log(-torch.expm1(-dt))
        self.dt_bias = nn.Parameter(inv_dt)  # fp32
        # A in [1, 16), stored as log (reference :398); fp32 like the reference
        A = torch.empty(self.nheads_local, dtype=torch.float32).uniform_(1, 16)
        self.A_log = nn.Parameter(torch.log(A))
        self.D = nn.Parameter(torch.ones(self.nheads_local, dtype=torch.float32))
        for p in (self.dt_bias, self.A_log, self.D):
            p.tensor_parallel = True
            p.partition_dim = 0

        self.norm = GatedRMSNorm(self.d_inner_local, self.ngroups_local,
                                 config.layernorm_epsilon, pdtype)
        self.out_proj = RowParallelLinear(
            self.d_inner, self.d_model, config=config, bias=config.add_linear_bias,
            input_is_parallel=True, skip_bias_add=False,
        )

    # ---- helpers -------------------------------------------------------

    def _conv(self, xBC: torch.Tensor) -> torch.Tensor:
        """Depthwise causal conv + SiLU over the sequence (fused HIP kernel
        K14 on GPU; MIOpen F.conv1d fallback). xBC: [b, l, conv_dim]."""
        from megatron_amd import ops

        return ops.causal_conv1d_silu(xBC, self.conv_weight, self.conv_bias)

    def allocate_inference_state(self, batch: int, device, dtype) -> dict:
        return {
            "conv": torch.zeros(batch, self.conv_dim_local, self.d_conv - 1, device=device, dtype=dtype),
            "ssm": torch.zeros(batch, self.nheads_local, self.headdim, self.d_state,
                               device=device, dtype=torch.float32),
        }

    # ---- forward -------------------------------------------------------

    def forward(self, hidden_states: torch.Tensor, inference_state: Optional[dict] = None):
        """hidden_states: [s, b, h] (sbh, like the attention path)."""
        # CP: the scan is sequential over the sequence, so CP shards are
        # gathered to the full sequence, scanned, and re-sliced (reference
        # mamba_context_parallel.py; this is the correct redundant-compute
        # baseline — head-split CP is the round-2 optimization)
        cp = G.get_context_parallel_world_size()
        cp_mode = getattr(self.config, "cp_comm_type", "p2p")
        # head-split CP (reference MambaContextParallel intent, Ulysses-style):
        # a2a the (x, B, C, dt) streams so each rank scans the FULL sequence
        # for 1/cp of the heads/groups — no redundant compute; z stays
        # sequence-sharded (it only gates the output after the inverse a2a).
        headsplit = (cp > 1 and inference_state is None
                     and self.nheads_local % cp == 0 and self.ngroups_local % cp == 0)
        if cp > 1 and inference_state is None and not headsplit:
            from megatron_amd.parallel.context_parallel import gather_cp_sequence

            hidden_states = gather_cp_sequence(hidden_states, seq_dim=0, mode=cp_mode)
        s, b, _ = hidden_states.shape
        zxbcdt, _ = self.in_proj(hidden_states)  # [s, b, d_proj/tp]
        z, x, B, C, dt = torch.split(zxbcdt, self.split_sizes, dim=-1)

        if inference_state is not None and s == 1:
            return self._step(z, x, B, C, dt, inference_state)

        if headsplit:
            return self._forward_headsplit_cp(z, x, B, C, dt, cp, cp_mode)

        xBC = torch.cat([x, B, C], dim=-1).permute(1, 0, 2)  # [b, l, conv_dim]
        if inference_state is not None:
            # prefill: remember the tail of the conv window
            tail = xBC.transpose(1, 2)[..., -(self.d_conv - 1):]
            pad = self.d_conv - 1 - tail.shape[-1]
            if pad > 0:
                tail = F.pad(tail, (pad, 0))
            inference_state["conv"].copy_(tail)
        xBC = self._conv(xBC)
        x, B, C = torch.split(
            xBC, [self.d_inner_local, self.ngroups_local * self.d_state,
                  self.ngroups_local * self.d_state], dim=-1,
        )
        x = x.view(b, s, self.nheads_local, self.headdim)
        B = B.view(b, s, self.ngroups_local, self.d_state)
        C = C.view(b, s, self.ngroups_local, self.d_state)
        dtv = F.softplus(dt.permute(1, 0, 2).float() + self.dt_bias)  # [b, l, h]
        A = -torch.exp(self.A_log)

        if inference_state is not None:
            y, final = ssd_chunked_scan(x, dtv.to(x.dtype), A, B, C, D=self.D,
                                        chunk_size=self.chunk_size, return_final_state=True)
            inference_state["ssm"].copy_(final)
        else:
            y = ssd_chunked_scan(x, dtv.to(x.dtype), A, B, C, D=self.D, chunk_size=self.chunk_size)

        y = y.reshape(b, s, self.d_inner_local).permute(1, 0, 2)  # [s, b, d_inner/tp]
        y = self.norm(y, z)
        out, _ = self.out_proj(y)
        if cp > 1 and inference_state is None:
            from megatron_amd.parallel.context_parallel import slice_for_cp_rank

            out = slice_for_cp_rank(out, G.get_context_parallel_rank(), cp,
                                    seq_dim=0, mode=cp_mode)
        return out

    def _headsplit_indices(self, cp: int, r: int, device):
        """Channel indices of CP chunk r inside the concatenated [x|B|C]
        conv layout (each component is chunked contiguously)."""
        dlc = self.d_inner_local // cp
        gsc = self.ngroups_local * self.d_state // cp
        xs = torch.arange(r * dlc, (r + 1) * dlc, device=device)
        bs = self.d_inner_local + torch.arange(r * gsc, (r + 1) * gsc, device=device)
        cs = self.d_inner_local + self.ngroups_local * self.d_state + torch.arange(
            r * gsc, (r + 1) * gsc, device=device)
        return torch.cat([xs, bs, cs])

    def _forward_headsplit_cp(self, z, x, B, C, dt, cp: int, cp_mode: str):
        from megatron_amd.parallel.context_parallel import (
            cp_channel_to_seq,
            cp_seq_to_channel,
        )

        r = G.get_context_parallel_rank()
        device = x.device
        # channel permutation so equal a2a chunks carry (x_r | B_r | C_r)
        perm = torch.cat([self._headsplit_indices(cp, rr, device) for rr in range(cp)])
        xBC = torch.cat([x, B, C], dim=-1).index_select(-1, perm)
        xBC = cp_seq_to_channel(xBC, mode=cp_mode)          # [s_full, b, conv_dim/cp]
        dtf = cp_seq_to_channel(dt, mode=cp_mode)           # [s_full, b, nheads/cp]
        s, b = xBC.shape[0], xBC.shape[1]

        # sliced per-channel weights for this rank's chunk
        my_idx = self._headsplit_indices(cp, r, device)
        conv_w = self.conv_weight.index_select(0, my_idx)
        conv_b = self.conv_bias.index_select(0, my_idx)
        from megatron_amd import ops as _ops

        xBCb = xBC.permute(1, 0, 2)                         # [b, l, conv_dim/cp]
        xBCb = _ops.causal_conv1d_silu(xBCb, conv_w, conv_b)

        nh, ng = self.nheads_local // cp, self.ngroups_local // cp
        xs, Bs, Cs = torch.split(
            xBCb, [nh * self.headdim, ng * self.d_state, ng * self.d_state], dim=-1)
        xs = xs.view(b, s, nh, self.headdim)
        Bs = Bs.view(b, s, ng, self.d_state)
        Cs = Cs.view(b, s, ng, self.d_state)
        hsel = torch.arange(r * nh, (r + 1) * nh, device=device)
        dtv = F.softplus(dtf.permute(1, 0, 2).float() + self.dt_bias.index_select(0, hsel))
        A = -torch.exp(self.A_log.index_select(0, hsel))
        y = ssd_chunked_scan(xs, dtv.to(xs.dtype), A, Bs, Cs,
                             D=self.D.index_select(0, hsel), chunk_size=self.chunk_size)
        y = y.reshape(b, s, nh * self.headdim).permute(1, 0, 2)  # [s_full, b, d_inner/(tp*cp)]
        y = cp_channel_to_seq(y, mode=cp_mode)              # [s_loc, b, d_inner/tp]
        y = self.norm(y, z)
        out, _ = self.out_proj(y)
        return out

    def _step(self, z, x, B, C, dt, state: dict):
        """Single-token decode. All inputs [1, b, ·]."""
        b = x.shape[1]
        xBC = torch.cat([x, B, C], dim=-1)[0]  # [b, conv_dim]
        conv = state["conv"]  # [b, conv_dim, d_conv-1]
        window = torch.cat([conv, xBC.unsqueeze(-1)], dim=-1)  # [b, conv_dim, d_conv]
        state["conv"].copy_(window[..., 1:])
        xBC = F.silu((window * self.conv_weight.squeeze(1)).sum(-1) + self.conv_bias)
        xs, Bs, Cs = torch.split(
            xBC, [self.d_inner_local, self.ngroups_local * self.d_state,
                  self.ngroups_local * self.d_state], dim=-1,
        )
        xs = xs.view(b, self.nheads_local, self.headdim)
        Bs = Bs.view(b, self.ngroups_local, self.d_state)
        Cs = Cs.view(b, self.ngroups_local, self.d_state)
        dts = F.softplus(dt[0].float() + self.dt_bias)  # [b, h]
        A = -torch.exp(self.A_log)
        y, _ = ssd_step(xs, dts.to(xs.dtype), A, Bs, Cs, state["ssm"], D=self.D)
        y = y.reshape(1, b, self.d_inner_local)
        y = self.norm(y, z)
        out, _ = self.out_proj(y)
        return out
