"""Configuration dataclasses.

Capability analog of the reference's TransformerConfig /
ModelParallelConfig / OptimizerConfig / DistributedDataParallelConfig
(megatron/core/transformer/transformer_config.py,
 megatron/core/model_parallel_config.py,
 megatron/core/optimizer/optimizer_config.py,
 megatron/core/distributed/distributed_data_parallel_config.py) —
re-designed as a lean, MI355X-first surface: only fields that change
behavior on a single 8-GPU xGMI node are first-class.
"""

from __future__ import annotations

import dataclasses
from dataclasses import dataclass, field
from typing import Callable, Optional, List

import torch


@dataclass
class ParallelConfig:
    """Process-grid shape.  World is factored rank = tp-fastest order
    ``tp -> cp -> dp -> pp`` (reference: parallel_state.py:601
    ``initialize_model_parallel`` default order 'tp-cp-ep-dp-pp')."""

    tensor_parallel_size: int = 1
    pipeline_parallel_size: int = 1
    context_parallel_size: int = 1
    cp_comm_type: str = "p2p"  # ring (p2p) or Ulysses head-scatter (a2a)
    mtp_num_layers: int = 0  # multi-token prediction depths (0 = off)
    causal_attention: bool = True  # False = bidirectional encoder (BERT)
    # multi-latent attention (DeepSeek): low-rank q/kv + decoupled rope
    multi_latent_attention: bool = False
    q_lora_rank: Optional[int] = None
    kv_lora_rank: int = 512
    qk_nope_head_dim: int = 64
    qk_rope_head_dim: int = 64
    v_head_dim: int = 128
    mtp_loss_scaling_factor: float = 0.1
    expert_parallel_size: int = 1
    virtual_pipeline_parallel_size: Optional[int] = None
    # uneven PP splits (reference --decoder-first/last-pipeline-num-layers):
    # give the embedding/loss stages fewer transformer layers
    num_layers_in_first_pipeline_stage: Optional[int] = None
    num_layers_in_last_pipeline_stage: Optional[int] = None
    sequence_parallel: bool = False
    fp8: object = None  # None | 'hybrid' | 'e4m3' (K15)
    fp8_amax_history_len: int = 16
    fp8_margin: int = 0
    # expert tensor parallel size (defaults to tensor_parallel_size)
    expert_tensor_parallel_size: Optional[int] = None


@dataclass
class TransformerConfig(ParallelConfig):
    """Model architecture + numerics config."""

    # ---- architecture ----
    num_layers: int = 2
    hidden_size: int = 64
    ffn_hidden_size: Optional[int] = None
    num_attention_heads: int = 4
    num_query_groups: Optional[int] = None  # GQA; None -> MHA
    kv_channels: Optional[int] = None
    vocab_size: int = 128
    max_position_embeddings: int = 4096
    # 'rmsnorm' (llama) or 'layernorm'
    normalization: str = "rmsnorm"
    layernorm_epsilon: float = 1e-5
    # 'swiglu' (llama), 'gelu', 'squared_relu'
    activation: str = "swiglu"
    add_linear_bias: bool = False
    add_qkv_bias: bool = False  # bias only on the fused QKV projection (Qwen2-style)  # llama-style: no bias anywhere
    untie_embeddings_and_output_weights: bool = True
    position_embedding_type: str = "rope"  # 'rope' | 'learned' | 'relative' | 'none'
    relative_attention_num_buckets: int = 32   # T5 relative-position bias
    relative_attention_max_distance: int = 128
    rotary_base: float = 500000.0  # llama-3 default
    rotary_percent: float = 1.0
    # None or {'type': 'llama3'|'linear', 'factor': ..., 'low_freq_factor': ...,
    # 'high_freq_factor': ..., 'original_max_position_embeddings': ...}
    rope_scaling: Optional[dict] = None
    label_smoothing: float = 0.0
    overlap_moe_expert_parallel_comm: bool = False  # combined-1F1B co-schedule
    attention_dropout: float = 0.0
    hidden_dropout: float = 0.0
    # sliding-window attention: None or window size (causal look-back)
    window_size: Optional[int] = None
    # every k-th layer (0-indexed: layers with number % k == k-1) uses FULL
    # attention, others the sliding window (reference window_attn_skip_freq;
    # gemma-2/llama-4 style interleaved local/global). None -> all windowed.
    window_attn_skip_freq: Optional[int] = None
    qk_layernorm: bool = False
    softmax_scale: Optional[float] = None

    # ---- MoE ----
    num_experts: Optional[int] = None
    moe_router_topk: int = 2
    moe_ffn_hidden_size: Optional[int] = None
    moe_aux_loss_coeff: float = 0.0
    moe_z_loss_coeff: float = 0.0
    moe_router_score_function: str = "softmax"  # 'softmax' | 'sigmoid'
    moe_router_pre_softmax: bool = False
    moe_shared_expert_intermediate_size: Optional[int] = None
    moe_grouped_gemm: bool = True
    moe_token_dispatcher_type: str = "alltoall"  # 'alltoall' | 'allgather'
    moe_expert_capacity_factor: Optional[float] = None  # None -> dropless
    moe_token_drop_policy: str = "probs"  # 'probs' | 'position'
    moe_pad_expert_input_to_capacity: bool = False  # static expert shapes
    moe_router_dtype: str = "fp32"
    # DeepSeek node-limited routing (reference moe_utils.py:673): experts split
    # into num_groups; each token routes only within its best group_topk groups.
    moe_router_num_groups: Optional[int] = None
    moe_router_group_topk: Optional[int] = None
    moe_input_jitter_eps: Optional[float] = None
    moe_aux_loss_type: str = "aux"  # 'aux' (switch, batch-level) | 'seq_aux'
    # aux-loss-free balancing (reference finalize_model_grads.py:334): after
    # each step, expert_bias += rate * sign(mean_load - expert_load).
    moe_router_enable_expert_bias: bool = False
    # benchmark/debug: force a perfectly balanced round-robin routing
    moe_router_force_load_balancing: bool = False
    moe_router_bias_update_rate: float = 1e-3
    # MuonClip: clip attention-logit growth by rescaling q/k weights after
    # each step (reference optimizer/qk_clip.py); None disables tracking.
    qk_clip_threshold: Optional[float] = None
    # layer frequency: 1 = every layer is MoE, k = every k-th layer
    moe_layer_freq: int = 1

    # ---- multi-token prediction ----
    mtp_num_layers: int = 0
    mtp_loss_scaling_factor: float = 0.1

    # ---- Mamba / hybrid SSM (reference ssm/mamba_mixer.py:144) ----
    mamba_state_dim: int = 128
    mamba_head_dim: int = 64
    mamba_num_groups: int = 8
    mamba_d_conv: int = 4
    mamba_expand: int = 2
    mamba_chunk_size: int = 128
    # hybrid layer allocation: explicit pattern ("M"/"*"/"-") wins over ratios
    hybrid_override_pattern: str = ""
    hybrid_attention_ratio: float = 0.0
    hybrid_mlp_ratio: float = 0.0

    # ---- numerics ----
    params_dtype: torch.dtype = torch.float32
    bf16: bool = False
    fp16: bool = False
    fp8: bool = False  # fp8 GEMM path (CDNA4 e4m3fn) — later phase
    attention_softmax_in_fp32: bool = True
    init_method_std: float = 0.02
    # activation recompute: None | 'full' | 'selective'
    recompute_granularity: Optional[str] = None
    recompute_num_layers: Optional[int] = None
    # activation CPU offloading (reference cpu_offloading): saved activations
    # of the first N layers live in pinned host memory between fwd and bwd
    # (288 GB HBM3E rarely needs it — this is for >seq-len-stretch cases)
    # selective-recompute module list (reference --recompute-modules):
    # "core_attn" (default), "mlp", "moe" — module-level rng-replay recompute
    recompute_modules: Optional[List[str]] = None
    activation_cpu_offload: bool = False
    activation_offload_layers: Optional[int] = None
    # distribute saved activations over TP group when recomputing
    distribute_saved_activations: bool = False

    # ---- execution ----
    # overlap TP dgrad all-reduce with wgrad GEMM (reference layers.py:622)
    async_tensor_model_parallel_allreduce: bool = True
    # SP forward AG as a chunked ring overlapped with the GEMM (TE
    # userbuffers ub_overlap_ag analog, parallel/overlap.py)
    tp_comm_overlap: bool = False
    gradient_accumulation_fusion: bool = True
    persist_layer_norm: bool = True
    deterministic_mode: bool = False
    cross_entropy_fusion: bool = True
    # hook points used by pipeline schedules (reference
    # model_parallel_config.py:211-223 three-hook contract)
    no_sync_func: Optional[Callable] = None
    grad_sync_func: Optional[Callable] = None
    param_sync_func: Optional[Callable] = None
    finalize_model_grads_func: Optional[Callable] = None
    # pipeline schedule knobs
    microbatch_group_size_per_vp_stage: Optional[int] = None
    overlap_p2p_comm: bool = True
    batch_p2p_comm: bool = False
    variable_seq_lengths: bool = False
    pipeline_dtype: Optional[torch.dtype] = None
    grad_scale_func: Optional[Callable] = None
    enable_autocast: bool = False
    autocast_dtype: Optional[torch.dtype] = None
    num_microbatches_with_partial_activation_checkpoints: Optional[int] = None
    deallocate_pipeline_outputs: bool = True
    defer_embedding_wgrad_compute: bool = False
    calculate_per_token_loss: bool = False

    timers: Optional[object] = None

    def __post_init__(self):
        if self.ffn_hidden_size is None:
            self.ffn_hidden_size = 4 * self.hidden_size
        if self.kv_channels is None:
            assert self.hidden_size % self.num_attention_heads == 0
            self.kv_channels = self.hidden_size // self.num_attention_heads
        if self.num_query_groups is None:
            self.num_query_groups = self.num_attention_heads
        if self.expert_tensor_parallel_size is None:
            self.expert_tensor_parallel_size = self.tensor_parallel_size
        if self.moe_ffn_hidden_size is None:
            self.moe_ffn_hidden_size = self.ffn_hidden_size
        if self.bf16:
            self.params_dtype = torch.bfloat16
        if self.fp16:
            self.params_dtype = torch.float16
        if self.pipeline_dtype is None:
            self.pipeline_dtype = self.params_dtype
        assert self.num_attention_heads % self.num_query_groups == 0
        if self.tensor_parallel_size > 1:
            assert self.num_attention_heads % self.tensor_parallel_size == 0
            assert (
                self.num_query_groups % self.tensor_parallel_size == 0
                or self.tensor_parallel_size % self.num_query_groups == 0
            )
        if self.sequence_parallel:
            assert self.tensor_parallel_size > 1 or True  # allowed; no-op at tp=1
        if self.num_experts is not None:
            assert self.num_experts % self.expert_parallel_size == 0

    def replace(self, **kw) -> "TransformerConfig":
        return dataclasses.replace(self, **kw)


@dataclass
class OptimizerConfig:
    """Analog of the reference OptimizerConfig (optimizer/optimizer_config.py)."""

    optimizer: str = "adam"  # 'adam' | 'muon' | 'sgd'
    sgd_momentum: float = 0.9
    # Muon (reference optimizer/muon.py): orthogonalized momentum for 2-D weights
    muon_momentum: float = 0.95
    muon_ns_steps: int = 5
    # keep fp32 main params + Adam moments in pinned host memory
    # (reference optimizer/cpu_offloading/ HybridDeviceOptimizer)
    optimizer_cpu_offload: bool = False
    lr: float = 1e-4
    min_lr: float = 0.0
    # separate lr for input/output embeddings (reference --decoupled-lr)
    decoupled_lr: Optional[float] = None
    decoupled_min_lr: Optional[float] = None
    weight_decay: float = 0.1
    adam_beta1: float = 0.9
    adam_beta2: float = 0.95
    adam_eps: float = 1e-8
    clip_grad: float = 1.0
    # mixed precision
    bf16: bool = False
    fp16: bool = False
    params_dtype: torch.dtype = torch.float32
    loss_scale: Optional[float] = None  # static; None -> dynamic when fp16
    initial_loss_scale: float = 2**32
    min_loss_scale: float = 1.0
    loss_scale_window: int = 1000
    hysteresis: int = 2
    # ZeRO-1 distributed optimizer
    use_distributed_optimizer: bool = False
    # >1: shard optimizer state over dp_cp/N-rank sub-groups and replicate
    # across the N instances (reference num_distributed_optimizer_instances):
    # smaller all-gather domains at large DP
    num_distributed_optimizer_instances: int = 1
    overlap_param_gather: bool = False
    # lr schedule
    lr_decay_style: str = "cosine"  # 'constant' | 'linear' | 'cosine' | 'wsd'
    lr_warmup_iters: int = 0
    lr_decay_iters: Optional[int] = None
    lr_wsd_decay_iters: Optional[int] = None
    start_weight_decay: Optional[float] = None
    end_weight_decay: Optional[float] = None
    weight_decay_incr_style: str = "constant"


@dataclass
class DDPConfig:
    """Analog of DistributedDataParallelConfig."""

    grad_reduce_in_fp32: bool = False
    overlap_grad_reduce: bool = True
    use_distributed_optimizer: bool = False
    # >1: shard optimizer state over dp_cp/N-rank sub-groups and replicate
    # across the N instances (reference num_distributed_optimizer_instances):
    # smaller all-gather domains at large DP
    num_distributed_optimizer_instances: int = 1
    bucket_size: Optional[int] = 40_000_000  # elements per bucket target
    average_in_collective: bool = True
    check_for_nan_in_grad: bool = False
    # allocate grad/param buffers in an RCCL-registered MemPool (N2)
    use_rccl_registered_buffers: bool = False
    align_param_gather: bool = False
