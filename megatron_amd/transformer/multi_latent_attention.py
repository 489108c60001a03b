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

    def forward(self, hidden_states, rotary_freqs=None, attention_mask=None,
                inference_context=None, packed_seq_params=None):
        assert inference_context is None, "MLA KV-cache (absorbed) path: round 2"
        assert packed_seq_params is None, "MLA packed (THD) path: round 2"
        s, b = hidden_states.shape[0], hidden_states.shape[1]
        if self.linear_q_down is not None:
            q, _ = self.linear_q_up(self.q_norm(self.linear_q_down(hidden_states)))
        else:
            q, _ = self.linear_q_up(hidden_states)
        q = q.view(s, b, self.nh, self.dqk)
        q_nope, q_rope = torch.split(q, [self.nope, self.rope], dim=3)

        down = self.linear_kv_down(hidden_states)  # [s, b, lora + rope]
        c_kv, k_rope = torch.split(down, [self.config.kv_lora_rank, self.rope], dim=2)
        kv, _ = self.linear_kv_up(self.kv_norm(c_kv))
        kv = kv.view(s, b, self.nh, self.nope + self.dv)
        k_nope, v = torch.split(kv, [self.nope, self.dv], dim=3)

        freqs = self._rope_freqs(s, hidden_states.device)
        q_rope = ops.rope_apply(q_rope.contiguous(), freqs)
        k_rope = ops.rope_apply(k_rope.view(s, b, 1, self.rope), freqs)

        qf = torch.cat([q_nope, q_rope], dim=3)
        kf = torch.cat([k_nope, k_rope.expand(s, b, self.nh, self.rope)], dim=3)
        if self.dqk in (64, 128) and self.dv == self.dqk:
            core = ops.flash_attention(qf, kf.contiguous(), v.contiguous(),
                                       causal=self.config.causal_attention,
                                       scale=self.softmax_scale)
        else:
            core = ref.attention(qf, kf, v, causal=self.config.causal_attention,
                                 scale=self.softmax_scale)
        out, _ = self.linear_proj(core.reshape(s, b, self.nh * self.dv))
        return out
