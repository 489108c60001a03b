"""Theoretical memory report (reference training/theoretical_memory_usage.py):
parameter / optimizer-state / activation estimates per GPU, printed at startup
so OOMs are predictable before the first step. Sized against 288 GB HBM3E.
"""

from __future__ import annotations

MI355X_HBM_GB = 288


def num_parameters(cfg) -> int:
    h, L, v = cfg.hidden_size, cfg.num_layers, cfg.vocab_size
    ffn = cfg.ffn_hidden_size
    ng = cfg.num_query_groups or cfg.num_attention_heads
    d = cfg.kv_channels
    qkv = h * (cfg.num_attention_heads + 2 * ng) * d
    proj = cfg.num_attention_heads * d * h
    if cfg.num_experts:
        mlp = cfg.num_experts * 3 * h * (cfg.moe_ffn_hidden_size or ffn) + h * cfg.num_experts
    else:
        mlp = 3 * h * ffn if cfg.activation in ("swiglu", "geglu") else 2 * h * ffn
    per_layer = qkv + proj + mlp + 2 * h
    emb = v * h * (2 if cfg.untie_embeddings_and_output_weights else 1)
    return L * per_layer + emb + h


def report(cfg, micro_batch_size: int, num_microbatches: int, dp_size: int = 1,
           use_distributed_optimizer: bool = True, recompute: bool = False) -> dict:
    """Returns (and the caller prints) the per-GPU memory model in GB."""
    n = num_parameters(cfg)
    tp = cfg.tensor_parallel_size
    pp = cfg.pipeline_parallel_size
    n_shard = n / tp / pp  # uniform approximation
    bytes_param = 2 * n_shard                      # bf16 weights
    bytes_grad = 4 * n_shard                       # fp32 main grads
    # distributed optimizer: fp32 master + 2 Adam moments sharded over DP
    opt_denom = dp_size if use_distributed_optimizer else 1
    bytes_opt = 12 * n_shard / opt_denom
    s, b, h = cfg.max_position_embeddings, micro_batch_size, cfg.hidden_size
    s = getattr(cfg, "seq_length", None) or s
    ffn = cfg.ffn_hidden_size
    # per-layer bf16 activations (flash attention: no s^2 term)
    act_per_layer = s * b * (2 * h * 6 + 2 * (2 * ffn) + 2 * ffn) / tp
    if recompute:
        act_per_layer = s * b * 2 * h  # boundary only
    acts = act_per_layer * cfg.num_layers / pp
    in_flight = min(num_microbatches, pp)
    out = {
        "params_b": n / 1e9,
        "weights_gb": bytes_param / 2**30,
        "grads_gb": bytes_grad / 2**30,
        "optimizer_gb": bytes_opt / 2**30,
        "activations_gb": acts * in_flight / 2**30,
    }
    out["total_gb"] = sum(v for k, v in out.items() if k.endswith("_gb"))
    out["hbm_gb"] = MI355X_HBM_GB
    return out


def format_report(r: dict) -> str:
    return (f"theoretical memory per GPU: weights {r['weights_gb']:.1f} + grads "
            f"{r['grads_gb']:.1f} + optimizer {r['optimizer_gb']:.1f} + activations "
            f"{r['activations_gb']:.1f} = {r['total_gb']:.1f} GB of {r['hbm_gb']} GB "
            f"({r['params_b']:.2f}B params)")
