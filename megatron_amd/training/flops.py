"""FLOP accounting for throughput/MFU reporting.

Same metric definition as reference megatron/training/training.py:391
(num_floating_point_operations): GEMM flops 2*M*N*K, backward = 2x forward,
causal attention halved.  MFU denominator for MI355X is the DENSE bf16 MFMA
peak (2.5 PFLOP/s per GPU — AMD's 5 PF figure includes 2:1 sparsity and is
never used here).
"""

from __future__ import annotations

MI355X_BF16_DENSE_PEAK_TFLOPS = 2500.0  # per GPU, dense (no sparsity)


def num_floating_point_operations(config, batch_size: int, seq_len: int) -> float:
    """Total fwd+bwd FLOPs for one global batch."""
    h = config.hidden_size
    L = config.num_layers
    f = config.ffn_hidden_size
    V = config.vocab_size
    kv = config.num_query_groups * config.kv_channels
    hq = config.num_attention_heads * config.kv_channels
    tokens = batch_size * seq_len

    gated = config.activation in ("swiglu", "geglu")
    per_token_layer = (
        2 * h * (hq + 2 * kv)          # qkv proj
        + 2 * hq * h                   # out proj
        + 2 * 2 * hq * (seq_len / 2)   # QK^T + PV, causal-halved
        + 2 * h * f * (3 if gated else 2)  # mlp
    )
    if config.num_experts is not None:
        fm = config.moe_ffn_hidden_size
        moe_mlp = 2 * h * fm * (3 if gated else 2) * config.moe_router_topk
        shared = 0
        if config.moe_shared_expert_intermediate_size:
            shared = 2 * h * config.moe_shared_expert_intermediate_size * 3
        freq = config.moe_layer_freq
        if isinstance(freq, (list, tuple)):
            n_moe = sum(bool(freq[i % len(freq)]) for i in range(L))
        else:
            n_moe = L // freq
        per_token_layer = per_token_layer - 2 * h * f * (3 if gated else 2) * (n_moe / L)
        per_token_layer += (moe_mlp + shared) * (n_moe / L)
    fwd = tokens * (L * per_token_layer + 2 * h * V)
    return 3.0 * fwd  # fwd + bwd(2x)


def tflops_per_gpu(config, batch_size: int, seq_len: int, iter_time_s: float, n_gpus: int) -> float:
    return num_floating_point_operations(config, batch_size, seq_len) / iter_time_s / n_gpus / 1e12


def mfu(config, batch_size: int, seq_len: int, iter_time_s: float, n_gpus: int) -> float:
    return tflops_per_gpu(config, batch_size, seq_len, iter_time_s, n_gpus) / MI355X_BF16_DENSE_PEAK_TFLOPS
