"""Multi-Latent Attention (DeepSeek-style).

Capability analog of reference megatron/core/transformer/
multi_latent_attention.py (:130 MLA, :471 MLASelfAttention): low-rank KV
compression (W_dkv -> kv_lora_rank latent, up-projected per head), optional
low-rank Q, and a decoupled RoPE channel: q/k split into a no-position (nope)
part and a rotary (rope) part where k_rope is a single shared head.

Default dims (nope 64 + rope 64 = 128 qk, v 128) keep the concatenated head
dim inside the flash kernel's supported set; other dims fall back to the
torch reference attention (absorbed-matmul inference path: round 2).
"""

from __future__ import annotations

import torch
import torch.nn as nn

from megatron_amd import ops
from megatron_amd.ops import reference as ref
from megatron_amd.parallel import grid as G
from megatron_amd.parallel.layers import ColumnParallelLinear, RowParallelLinear
from megatron_amd.transformer.block import Norm


class MLASelfAttention(nn.Module):
    def __init__(self, config, layer_number: int = 0):
        super().__init__()
        self.config = config
        self.layer_number = layer_number
        tp = G.get_tensor_model_parallel_world_size()
        h = config.hidden_size
        self.nope = config.qk_nope_head_dim
        self.rope = config.qk_rope_head_dim
        self.dqk = self.nope + self.rope
        self.dv = config.v_head_dim
        assert config.num_attention_heads % tp == 0
        self.nh = config.num_attention_heads // tp

        if config.q_lora_rank:
            self.linear_q_down = nn.Linear(h, config.q_lora_rank, bias=False,
                                           dtype=config.params_dtype)
            self.q_norm = Norm(config, hidden_size=config.q_lora_rank)
            self.linear_q_up = ColumnParallelLinear(
                config.q_lora_rank, config.num_attention_heads * self.dqk,
                config=config, bias=False)
        else:
            self.linear_q_down = None
            self.linear_q_up = ColumnParallelLinear(
                h, config.num_attention_heads * self.dqk, config=config, bias=False)

        # W_dkv emits [kv latent | shared k_rope head] (replicated over TP)
        self.linear_kv_down = nn.Linear(h, config.kv_lora_rank + self.rope, bias=False,
                                        dtype=config.params_dtype)
        self.kv_norm = Norm(config, hidden_size=config.kv_lora_rank)
        self.linear_kv_up = ColumnParallelLinear(
            config.kv_lora_rank, config.num_attention_heads * (self.nope + self.dv),
            config=config, bias=False)
        self.linear_proj = RowParallelLinear(
            config.num_attention_heads * self.dv, h, config=config, bias=False)
        self.softmax_scale = config.softmax_scale or (1.0 / (self.dqk ** 0.5))
        self._freq_cache = {}

    def _rope_freqs(self, s: int, device):
        key = (s, str(device))
        if key not in self._freq_cache:
            self._freq_cache.clear()
            self._freq_cache[key] = ref.rope_freqs(s, self.rope, base=self.config.rotary_base,
                                                   device=device)
        return self._freq_cache[key]

    # -- absorbed-matmul inference path (reference :1231 FusedMLASelfAttention)
    #
    # The KV cache stores the LATENT per token: [normed c_kv (r) | roped
    # k_rope (rope)] — (r + rope) elements instead of nh*(dqk + dv).  At
    # attend time W_uk is absorbed into the query (q̃ = q_nope @ W_uk, per
    # head) so scores are taken directly against the latent, and W_uv is
    # applied after the probability-weighted latent sum.  Algebraically
    # identical to the training path: q_nope·(W_uk c) == (q_nope W_uk)·c and
    # Σ p (W_uv c) == W_uv (Σ p c).

    def _latent_cache(self, ctx, device, dtype):
        store = getattr(ctx, "mla_latent_cache", None)
        if store is None:
            store = {}
            ctx.mla_latent_cache = store
        if self.layer_number not in store:
            width = self.config.kv_lora_rank + self.rope
            store[self.layer_number] = torch.zeros(ctx.max_seq, ctx.max_batch, width,
                                                   dtype=dtype, device=device)
        return store[self.layer_number]

    def _uk_uv(self):
        w = self.linear_kv_up.weight  # [nh*(nope+dv), r]
        w = w.view(self.nh, self.nope + self.dv, -1)
        return w[:, : self.nope], w[:, self.nope:]  # [nh, nope, r], [nh, dv, r]

    def _latent_pool(self, ctx, device):
        """Paged latent pool mirroring the dynamic context's block structure
        ([num_blocks+1, block_size, r+rope], slot-indexed by the same block
        tables)."""
        store = getattr(ctx, "mla_latent_pool", None)
        if store is None:
            store = {}
            ctx.mla_latent_pool = store
        if self.layer_number not in store:
            width = self.config.kv_lora_rank + self.rope
            n_slots = ctx.k_cache[0].shape[0]
            store[self.layer_number] = torch.zeros(n_slots, ctx.block_size, width,
                                                   dtype=torch.float32, device=device)
        return store[self.layer_number]

    def _absorbed_attend_paged(self, ctx, q_nope, q_rope, c_kv, k_rope):
        """Dynamic (paged) twin of _absorbed_attend: latents live in a paged
        pool addressed by the engine's block tables."""
        s, b = q_nope.shape[0], q_nope.shape[1]
        r = self.config.kv_lora_rank
        bs = ctx.block_size
        W_uk, W_uv = self._uk_uv()
        q_abs = torch.einsum("sbhn,hnr->sbhr", q_nope.float(), W_uk.float())
        lat_q = torch.cat([q_abs, q_rope.float()], dim=-1)
        pool = self._latent_pool(ctx, q_nope.device).view(-1, r + self.rope)
        new_latent = torch.cat([c_kv.float(), k_rope.float()], dim=-1)  # [s,b,r+rope]

        if ctx._mode == "prefill":
            assert b == 1
            pos = torch.arange(ctx._prior_len, ctx._prior_len + s, device=q_nope.device)
            pool[ctx._slot_index(pos, ctx._prefill_table)] = new_latent[:, 0]
            L = ctx._prior_len + s
            all_slots = ctx._slot_index(torch.arange(L, device=q_nope.device), ctx._prefill_table)
            kv_lat = pool[all_slots].unsqueeze(1)  # [L, 1, r+rope]
            logits = torch.einsum("sbhr,lbr->sbhl", lat_q, kv_lat) * self.softmax_scale
            key_pos = torch.arange(L, device=q_nope.device)
            mask = key_pos.view(1, 1, 1, L) <= pos.view(s, 1, 1, 1)
            logits = logits.masked_fill(~mask, float("-inf"))
            probs = torch.softmax(logits, dim=-1)
            c_hat = torch.einsum("sbhl,lbr->sbhr", probs, kv_lat[..., :r])
        else:  # ragged decode over block tables
            lens = ctx._context_lens
            rows = torch.arange(b, device=q_nope.device)
            slots = (ctx._block_tables[rows, torch.div(lens, bs, rounding_mode="floor")] * bs
                     + lens % bs)
            pool[slots] = new_latent[0]
            new_lens = lens + 1
            if ctx._static:
                tables = ctx._block_tables
            else:
                nb = int(torch.div(new_lens.max() + bs - 1, bs, rounding_mode="floor"))
                tables = ctx._block_tables[:, :nb]
            slot_grid = (tables.unsqueeze(-1) * bs +
                         torch.arange(bs, device=q_nope.device).view(1, 1, bs)).reshape(b, -1)
            kv_lat = pool[slot_grid]  # [b, L, r+rope]
            L = kv_lat.shape[1]
            logits = torch.einsum("bhr,blr->bhl", lat_q[0], kv_lat) * self.softmax_scale
            key_pos = torch.arange(L, device=q_nope.device)
            mask = key_pos.view(1, 1, L) < new_lens.view(b, 1, 1)
            logits = logits.masked_fill(~mask, float("-inf"))
            probs = torch.softmax(logits, dim=-1)
            c_hat = torch.einsum("bhl,blr->bhr", probs, kv_lat[..., :r]).unsqueeze(0)
        out = torch.einsum("sbhr,hdr->sbhd", c_hat, W_uv.float())
        return out.to(q_nope.dtype)

    def _absorbed_attend(self, ctx, q_nope, q_rope, c_kv, k_rope):
        """q_nope [s,b,nh,nope], q_rope [s,b,nh,rope], c_kv (normed)
        [s,b,r], k_rope (roped) [s,b,rope] -> core out [s,b,nh,dv]."""
        s, b = q_nope.shape[0], q_nope.shape[1]
        r = self.config.kv_lora_rank
        W_uk, W_uv = self._uk_uv()
        q_abs = torch.einsum("sbhn,hnr->sbhr", q_nope.float(), W_uk.float())
        lat_q = torch.cat([q_abs, q_rope.float()], dim=-1)  # [s,b,nh,r+rope]

        cache = self._latent_cache(ctx, q_nope.device, torch.float32)
        lens = ctx.context_lens[:b]
        new_latent = torch.cat([c_kv.float(), k_rope.float()], dim=-1)  # [s,b,r+rope]
        if s > 1:  # prefill (uniform offset)
            off = int(lens[0])
            cache[off:off + s, :b] = new_latent
            L = off + s
            pos_q = off + torch.arange(s, device=q_nope.device)
        else:  # ragged decode: scatter at each row's length
            ar = torch.arange(b, device=q_nope.device)
            cache[lens, ar] = new_latent[0]
            L = int((lens + 1).max())
            pos_q = lens.view(1, -1)  # [1, b]
        kv_lat = cache[:L, :b]  # [L, b, r+rope]
        logits = torch.einsum("sbhr,lbr->sbhl", lat_q, kv_lat) * self.softmax_scale
        key_pos = torch.arange(L, device=q_nope.device)
        if pos_q.dim() == 1:
            mask = key_pos.view(1, 1, L) <= pos_q.view(s, 1, 1)  # [s,1,L]
        else:
            mask = key_pos.view(1, 1, L) <= pos_q.view(1, b, 1)  # [1,b,L]
        logits = logits.masked_fill(~mask.unsqueeze(2), float("-inf"))
        probs = torch.softmax(logits, dim=-1)
        c_hat = torch.einsum("sbhl,lbr->sbhr", probs, kv_lat[..., :r])
        out = torch.einsum("sbhr,hdr->sbhd", c_hat, W_uv.float())
        return out.to(q_nope.dtype)

    def forward(self, hidden_states, rotary_freqs=None, attention_mask=None,
                inference_context=None, packed_seq_params=None):
        s, b = hidden_states.shape[0], hidden_states.shape[1]
        if self.linear_q_down is not None:
            q, _ = self.linear_q_up(self.q_norm(self.linear_q_down(hidden_states)))
        else:
            q, _ = self.linear_q_up(hidden_states)
        q = q.view(s, b, self.nh, self.dqk)
        q_nope, q_rope = torch.split(q, [self.nope, self.rope], dim=3)

        down = self.linear_kv_down(hidden_states)  # [s, b, lora + rope]
        c_kv, k_rope = torch.split(down, [self.config.kv_lora_rank, self.rope], dim=2)

        if inference_context is not None:
            max_pos = getattr(inference_context, "max_seq", self.config.max_position_embeddings)
            table = self._rope_freqs(max_pos, hidden_states.device)
            pos = inference_context.rope_positions(s)  # [s] or [1, b]
            fr = table[pos]
            if fr.dim() == 2:  # [s, rope/2]
                q_rope = ops.rope_apply(q_rope.contiguous(), fr)
                k_rope = ops.rope_apply(k_rope.view(s, b, 1, self.rope), fr)
            else:  # ragged decode: [1, b, rope/2]
                q_rope = ref.rope_apply_per_row(q_rope.contiguous(), fr)
                k_rope = ref.rope_apply_per_row(k_rope.view(s, b, 1, self.rope), fr)
            # NOTE: c_kv is cached post-norm; k_rope cached post-rope
            attend = (self._absorbed_attend_paged
                      if hasattr(inference_context, "block_size")
                      else self._absorbed_attend)
            core = attend(inference_context, q_nope, q_rope,
                          self.kv_norm(c_kv), k_rope.view(s, b, self.rope))
            out, _ = self.linear_proj(core.reshape(s, b, self.nh * self.dv))
            return out

        kv, _ = self.linear_kv_up(self.kv_norm(c_kv))
        kv = kv.view(s, b, self.nh, self.nope + self.dv)
        k_nope, v = torch.split(kv, [self.nope, self.dv], dim=3)

        cp = G.get_context_parallel_world_size() if G.grid_initialized() else 1
        if packed_seq_params is not None:
            # THD pack: per-document RoPE positions + block-diagonal attention
            table = self._rope_freqs(packed_seq_params.max_seqlen, hidden_states.device)
            freqs = table[packed_seq_params.positions().to(hidden_states.device)]
        elif cp > 1:
            # CP: this rank holds a zigzag (p2p) or contiguous (a2a) shard of
            # the global sequence; RoPE must use GLOBAL positions
            from megatron_amd.parallel.context_parallel import cp_rope_positions

            s_global = s * cp
            table = self._rope_freqs(s_global, hidden_states.device)
            pos = cp_rope_positions(s_global, G.get_context_parallel_rank(), cp,
                                    hidden_states.device, mode=self.config.cp_comm_type)
            freqs = table[pos]
        else:
            freqs = self._rope_freqs(s, hidden_states.device)
        q_rope = ops.rope_apply(q_rope.contiguous(), freqs)
        k_rope = ops.rope_apply(k_rope.view(s, b, 1, self.rope), freqs)

        qf = torch.cat([q_nope, q_rope], dim=3)
        kf = torch.cat([k_nope, k_rope.expand(s, b, self.nh, self.rope)], dim=3)
        if cp > 1 and packed_seq_params is None:
            from megatron_amd.parallel.context_parallel import ring_attention, ulysses_attention

            if self.config.cp_comm_type == "a2a":
                core = ulysses_attention(qf, kf.contiguous(), v.contiguous(),
                                         scale=self.softmax_scale)
            else:
                core = ring_attention(qf, kf.contiguous(), v.contiguous(),
                                      scale=self.softmax_scale)
            out, _ = self.linear_proj(core.reshape(s, b, self.nh * self.dv))
            return out
        if packed_seq_params is not None:
            core = ref.attention_varlen(
                qf, kf.contiguous(), v.contiguous(),
                packed_seq_params.cu_seqlens.to(qf.device),
                causal=self.config.causal_attention, scale=self.softmax_scale)
        elif self.dqk in (64, 128) and self.dv == self.dqk:
            core = ops.flash_attention(qf, kf.contiguous(), v.contiguous(),
                                       causal=self.config.causal_attention,
                                       scale=self.softmax_scale)
        else:
            core = ref.attention(qf, kf, v, causal=self.config.causal_attention,
                                 scale=self.softmax_scale)
        out, _ = self.linear_proj(core.reshape(s, b, self.nh * self.dv))
        return out
