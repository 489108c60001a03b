"""CLI argument surface.

Capability analog of reference megatron/training/arguments.py (:45
add_megatron_arguments, 36 groups / ~1,200 flags): the flags that change
behavior on an MI355X node, grouped the same way, with parse-time
cross-validation and a translation into the config dataclasses.
"""

from __future__ import annotations

import argparse

import torch

from megatron_amd.config import DDPConfig, OptimizerConfig, TransformerConfig


def build_arg_parser() -> argparse.ArgumentParser:
    p = argparse.ArgumentParser(description="megatron_amd pretraining", allow_abbrev=False)

    p.add_argument("--yaml-cfg", type=str, default=None,
                   help="yaml config file overlaid onto defaults (CLI flags win)")

    g = p.add_argument_group("model")
    g.add_argument("--num-layers", type=int, required=False, default=2)
    g.add_argument("--hidden-size", type=int, default=64)
    g.add_argument("--ffn-hidden-size", type=int, default=None)
    g.add_argument("--num-attention-heads", type=int, default=4)
    g.add_argument("--num-query-groups", type=int, default=None)
    g.add_argument("--kv-channels", type=int, default=None)
    g.add_argument("--vocab-size", type=int, default=128256)
    g.add_argument("--make-vocab-size-divisible-by", type=int, default=None,
                   help="pad the vocab so each TP shard is a multiple of this")
    g.add_argument("--max-position-embeddings", type=int, default=4096)
    g.add_argument("--normalization", choices=["rmsnorm", "layernorm"], default="rmsnorm")
    g.add_argument("--norm-epsilon", type=float, default=1e-5)
    g.add_argument("--swiglu", action="store_true", default=True)
    g.add_argument("--activation", choices=["swiglu", "geglu", "gelu", "squared_relu"], default="swiglu")
    g.add_argument("--add-linear-bias", action="store_true")
    g.add_argument("--untie-embeddings-and-output-weights", action="store_true", default=True)
    g.add_argument("--position-embedding-type", choices=["rope", "learned", "relative", "none"], default="rope")
    g.add_argument("--rotary-base", type=float, default=500000.0)
    g.add_argument("--rotary-percent", type=float, default=1.0)
    g.add_argument("--attention-dropout", type=float, default=0.0)
    g.add_argument("--hidden-dropout", type=float, default=0.0)
    g.add_argument("--window-size", type=int, default=None)
    g.add_argument("--qk-layernorm", action="store_true")
    g.add_argument("--qk-clip-threshold", type=float, default=None,
                   help="per-head qk-logit clip tracking threshold (muon/qk-clip)")
    g.add_argument("--rope-scaling", action="store_true",
                   help="llama-3.1-style rope frequency scaling")
    g.add_argument("--init-method-std", type=float, default=0.02)
    g.add_argument("--mtp-num-layers", type=int, default=0)
    g.add_argument("--multi-latent-attention", action="store_true")
    # Mamba / hybrid SSM (reference ssm/ CLI surface)
    g.add_argument("--mamba-state-dim", type=int, default=128)
    g.add_argument("--mamba-head-dim", type=int, default=64)
    g.add_argument("--mamba-num-groups", type=int, default=8)
    g.add_argument("--hybrid-override-pattern", type=str, default="")
    g.add_argument("--hybrid-attention-ratio", type=float, default=0.0)
    g.add_argument("--hybrid-mlp-ratio", type=float, default=0.0)
    g.add_argument("--q-lora-rank", type=int, default=None)
    g.add_argument("--kv-lora-rank", type=int, default=512)
    g.add_argument("--qk-nope-head-dim", type=int, default=64)
    g.add_argument("--qk-rope-head-dim", type=int, default=64)
    g.add_argument("--v-head-dim", type=int, default=128)
    g.add_argument("--mtp-loss-scaling-factor", type=float, default=0.1)

    g = p.add_argument_group("moe")
    g.add_argument("--num-experts", type=int, default=None)
    g.add_argument("--moe-router-topk", type=int, default=2)
    g.add_argument("--moe-ffn-hidden-size", type=int, default=None)
    g.add_argument("--moe-aux-loss-coeff", type=float, default=0.0)
    g.add_argument("--moe-z-loss-coeff", type=float, default=0.0)
    g.add_argument("--moe-router-score-function", choices=["softmax", "sigmoid"], default="softmax")
    g.add_argument("--moe-shared-expert-intermediate-size", type=int, default=None)
    g.add_argument("--moe-token-dispatcher-type", choices=["alltoall", "allgather"], default="alltoall")
    def _moe_freq(v):
        try:
            return int(v)
        except ValueError:
            import ast

            pat = ast.literal_eval(v)  # e.g. "[0]*1+[1]*3" forms are not allowed; pass a plain list
            assert isinstance(pat, (list, tuple)) and all(x in (0, 1) for x in pat)
            return list(pat)
    g.add_argument("--moe-router-force-load-balancing", action="store_true", default=False)
    g.add_argument("--moe-layer-freq", type=_moe_freq, default=1,
                   help="int N (every Nth layer is MoE) or a 0/1 list like [0,0,1,1]")
    g.add_argument("--moe-router-pre-softmax", action="store_true", default=False)
    g.add_argument("--moe-router-num-groups", type=int, default=None)
    g.add_argument("--moe-router-group-topk", type=int, default=None)
    g.add_argument("--moe-input-jitter-eps", type=float, default=None)
    g.add_argument("--moe-aux-loss-type", choices=["aux", "seq_aux"], default="aux")
    g.add_argument("--moe-router-enable-expert-bias", action="store_true", default=False)
    g.add_argument("--moe-router-bias-update-rate", type=float, default=1e-3)
    g.add_argument("--moe-expert-capacity-factor", type=float, default=None,
                   help="None = dropless; otherwise capacity = ceil(T*topk/E * factor)")
    g.add_argument("--moe-token-drop-policy", choices=["probs", "position"], default="probs")
    g.add_argument("--moe-pad-expert-input-to-capacity", action="store_true",
                   help="pad each expert's tokens to exactly capacity (static grouped-GEMM shapes)")

    g = p.add_argument_group("parallelism")
    g.add_argument("--tensor-model-parallel-size", "--tp", type=int, default=1)
    g.add_argument("--pipeline-model-parallel-size", "--pp", type=int, default=1)
    g.add_argument("--virtual-pipeline-model-parallel-size", "--vpp", type=int, default=None)
    g.add_argument("--context-parallel-size", "--cp", type=int, default=1)
    g.add_argument("--cp-comm-type", choices=["p2p", "a2a"], default="p2p")
    g.add_argument("--decoder-first-pipeline-num-layers", type=int, default=None,
                   dest="num_layers_in_first_pipeline_stage")
    g.add_argument("--decoder-last-pipeline-num-layers", type=int, default=None,
                   dest="num_layers_in_last_pipeline_stage")
    g.add_argument("--expert-model-parallel-size", "--ep", type=int, default=1)
    g.add_argument("--expert-tensor-parallel-size", type=int, default=None)
    g.add_argument("--sequence-parallel", action="store_true")
    g.add_argument("--nccl-communicator-config-path", type=str, default=None,
                   help="yaml of per-group RCCL knobs (min/max CTAs, stream priority)")
    g.add_argument("--tp-comm-overlap", action="store_true",
                   help="SP forward AG as a chunked ring overlapped with the GEMM")
    g.add_argument("--use-rccl-registered-buffers", action="store_true",
                   help="allocate DDP buffers in an RCCL-registered pool (zero-copy xGMI)")
    g.add_argument("--hip-graphs", action="store_true",
                   help="capture per-layer fwd/bwd hipGraphs (static shapes, dense layers)")
    g.add_argument("--overlap-moe-expert-parallel-comm", action="store_true",
                   help="combined-1F1B: layer-granular fwd/bwd co-schedule so MoE EP a2a overlaps compute")
    g.add_argument("--use-fsdp", action="store_true",
                   help="Megatron-FSDP-style per-layer param/grad sharding instead of DDP+ZeRO-1")

    g = p.add_argument_group("training")
    g.add_argument("--micro-batch-size", type=int, default=1)
    g.add_argument("--global-batch-size", type=int, default=None)
    g.add_argument("--rampup-batch-size", type=int, nargs=3, default=None,
                   metavar=("START", "INCREMENT", "SAMPLES"),
                   help="grow GBS from START by INCREMENT over SAMPLES consumed samples")
    g.add_argument("--seq-length", type=int, default=4096)
    g.add_argument("--reset-attention-mask", action="store_true", default=False,
                   help="document-boundary (EOD) attention reset: samples carry "
                        "cu_seqlens and train block-diagonal via the varlen path")
    g.add_argument("--reset-position-ids", action="store_true", default=False,
                   help="restart positions at each EOD (implied by "
                        "--reset-attention-mask on the varlen path)")
    g.add_argument("--eod-mask-loss", action="store_true", default=False)
    g.add_argument("--data-cache-path", type=str, default=None,
                   help="directory for dataset index caches (default: next to the data)")
    g.add_argument("--eod-id", type=int, default=None,
                   help="EOD token id for the reset/mask flags (default: tokenizer eod)")
    g.add_argument("--packed-sequences", action="store_true",
                   help="THD training: microbatch rows flattened into one packed stream "
                        "(block-diagonal attention, per-document RoPE restart)")
    g.add_argument("--train-iters", type=int, default=10)
    g.add_argument("--eval-interval", type=int, default=0)
    g.add_argument("--eval-iters", type=int, default=2)
    g.add_argument("--exit-interval", type=int, default=None)
    g.add_argument("--exit-signal-handler", action="store_true")
    g.add_argument("--exit-duration-in-mins", type=float, default=None)
    g.add_argument("--recompute-granularity", choices=["full", "selective"], default=None)
    g.add_argument("--recompute-num-layers", type=int, default=None)
    g.add_argument("--bf16", action="store_true")
    g.add_argument("--fp8-format", choices=["hybrid", "e4m3"], default=None)
    g.add_argument("--fp8-amax-history-len", type=int, default=16)
    g.add_argument("--fp8-margin", type=int, default=0)
    g.add_argument("--fp16", action="store_true")
    g.add_argument("--label-smoothing", type=float, default=0.0)
    g.add_argument("--seed", type=int, default=1234)
    g.add_argument("--deterministic-mode", action="store_true")

    g = p.add_argument_group("optimizer")
    g.add_argument("--optimizer", choices=["adam", "muon", "sgd"], default="adam")
    g.add_argument("--sgd-momentum", type=float, default=0.9)
    g.add_argument("--optimizer-cpu-offload", action="store_true", default=False)
    g.add_argument("--muon-momentum", type=float, default=0.95)
    g.add_argument("--lr", type=float, default=3e-4)
    g.add_argument("--min-lr", type=float, default=0.0)
    g.add_argument("--decoupled-lr", type=float, default=None)
    g.add_argument("--decoupled-min-lr", type=float, default=None)
    g.add_argument("--lr-decay-style", choices=["constant", "linear", "cosine", "wsd"], default="cosine")
    g.add_argument("--lr-warmup-iters", type=int, default=0)
    g.add_argument("--lr-decay-iters", type=int, default=None)
    g.add_argument("--weight-decay", type=float, default=0.1)
    g.add_argument("--adam-beta1", type=float, default=0.9)
    g.add_argument("--adam-beta2", type=float, default=0.95)
    g.add_argument("--adam-eps", type=float, default=1e-8)
    g.add_argument("--clip-grad", type=float, default=1.0)
    g.add_argument("--loss-scale", type=float, default=None)
    g.add_argument("--use-distributed-optimizer", action="store_true")
    g.add_argument("--overlap-param-gather", action="store_true")

    g = p.add_argument_group("ddp")
    g.add_argument("--overlap-grad-reduce", action="store_true", default=True)
    g.add_argument("--no-overlap-grad-reduce", dest="overlap_grad_reduce", action="store_false")
    g.add_argument("--accumulate-allreduce-grads-in-fp32", dest="grad_reduce_in_fp32",
                   action="store_true", default=True)
    g.add_argument("--bucket-size", type=int, default=40_000_000)
    g.add_argument("--check-for-nan-in-loss-and-grad", dest="check_for_nan_in_grad",
                   action="store_true", default=False)
    g.add_argument("--num-distributed-optimizer-instances", type=int, default=1)
    g.add_argument("--activation-cpu-offload", action="store_true", default=False)
    g.add_argument("--recompute-modules", nargs="+", default=None,
                   choices=["core_attn", "mlp", "moe"])
    g.add_argument("--activation-offload-layers", type=int, default=None)

    g = p.add_argument_group("checkpointing")
    g.add_argument("--save", type=str, default=None)
    g.add_argument("--load", type=str, default=None)
    g.add_argument("--save-interval", type=int, default=None)
    g.add_argument("--non-persistent-ckpt-dir", type=str, default=None,
                   help="node-local (SSD/ramdisk) fast-restart checkpoint tree")
    g.add_argument("--non-persistent-save-interval", type=int, default=None)
    g.add_argument("--async-save", action="store_true")
    g.add_argument("--no-load-rng", action="store_true")
    g.add_argument("--no-load-optim", action="store_true")
    g.add_argument("--finetune", action="store_true",
                   help="load model weights only; reset iteration, optimizer and schedule")

    g = p.add_argument_group("data")
    g.add_argument("--data-path", type=str, nargs="*", default=None)
    g.add_argument("--mock-data", action="store_true")
    g.add_argument("--tokenizer-type", type=str, default="NullTokenizer")
    g.add_argument("--tokenizer-model", type=str, default=None)
    g.add_argument("--split", type=str, default="969,30,1")
    g.add_argument("--num-workers", type=int, default=0)

    g = p.add_argument_group("logging")
    g.add_argument("--log-interval", type=int, default=1)
    g.add_argument("--log-throughput", action="store_true", default=True)
    g.add_argument("--log-timers", action="store_true")
    g.add_argument("--tensorboard-dir", type=str, default=None)
    g.add_argument("--log-memory", action="store_true")
    g.add_argument("--log-params-norm", action="store_true")
    g.add_argument("--log-num-zeros-in-grad", action="store_true")
    g.add_argument("--log-straggler", action="store_true")
    g.add_argument("--straggler-report-interval", type=int, default=10)
    g.add_argument("--straggler-ctrlr-port", type=int, default=None,
                   help="TCP port for runtime straggler-detection toggling (curl host:port)")
    g.add_argument("--use-wandb", action="store_true")
    g.add_argument("--wandb-project", type=str, default=None)
    g.add_argument("--check-weight-hash-across-dp-replicas-interval", type=int, default=None)
    g.add_argument("--rerun-mode", choices=["disabled", "validate_results"], default="disabled")
    g.add_argument("--fault-injection-type", choices=["crash", "hang", "nan_loss"], default=None)
    g.add_argument("--fault-injection-iteration", type=int, default=None)
    g.add_argument("--fault-injection-ranks", type=int, nargs="*", default=[])
    g.add_argument("--inprocess-restarts", type=int, default=0)
    g.add_argument("--log-energy", action="store_true")

    g = p.add_argument_group("profiling")
    g.add_argument("--profile", action="store_true")
    g.add_argument("--gpu-sniff-test", action="store_true",
                   help="pre-run GEMM health check on every visible GPU")
    g.add_argument("--profile-ranges", action="store_true",
                   help="emit rocTX/profiler range annotations")
    g.add_argument("--profile-step-start", type=int, default=3)
    g.add_argument("--profile-step-end", type=int, default=5)
    g.add_argument("--profile-dir", type=str, default="./torchprof")
    g.add_argument("--memory-snapshot-path", type=str, default=None,
                   help="dump a torch.cuda memory-history snapshot here at exit")

    return p


def validate_args(args) -> None:
    world = args.world_size
    mp = args.tensor_model_parallel_size * args.pipeline_model_parallel_size * args.context_parallel_size
    assert world % mp == 0, f"world {world} not divisible by tp*pp*cp={mp}"
    dp = world // mp
    if args.global_batch_size is None:
        args.global_batch_size = args.micro_batch_size * dp
    gbs_div = args.micro_batch_size * dp
    assert args.global_batch_size % gbs_div == 0, (
        f"global batch {args.global_batch_size} must divide by micro_batch*dp={gbs_div}"
    )
    args.num_microbatches = args.global_batch_size // gbs_div
    args.data_parallel_size = dp
    if args.make_vocab_size_divisible_by:
        from megatron_amd.tokenizers import pad_vocab_size

        args.vocab_size = pad_vocab_size(args.vocab_size, args.tensor_model_parallel_size,
                                         args.make_vocab_size_divisible_by)
    if args.fp16 and args.bf16:
        raise ValueError("choose one of --fp16 / --bf16")
    if args.sequence_parallel and args.tensor_model_parallel_size == 1:
        args.sequence_parallel = False
    if args.num_experts is not None:
        assert args.num_experts % args.expert_model_parallel_size == 0
        if isinstance(args.moe_layer_freq, (list, tuple)):
            assert len(args.moe_layer_freq) <= args.num_layers, (
                "--moe-layer-freq pattern longer than the layer count")
    if getattr(args, "reset_attention_mask", False) or getattr(args, "eod_mask_loss", False):
        assert args.mock_data or args.eod_id is not None or args.tokenizer_type, (
            "--reset-attention-mask/--eod-mask-loss need --eod-id or a tokenizer")
    inst = getattr(args, "num_distributed_optimizer_instances", 1)
    if inst > 1:
        assert args.use_distributed_optimizer, (
            "--num-distributed-optimizer-instances needs --use-distributed-optimizer")
        assert (args.data_parallel_size * args.context_parallel_size) % inst == 0, (
            "dp_cp must divide evenly into optimizer instances")


def configs_from_args(args):
    cfg = TransformerConfig(
        num_layers=args.num_layers,
        hidden_size=args.hidden_size,
        ffn_hidden_size=args.ffn_hidden_size,
        num_attention_heads=args.num_attention_heads,
        num_query_groups=args.num_query_groups,
        kv_channels=args.kv_channels,
        vocab_size=args.vocab_size,
        max_position_embeddings=args.max_position_embeddings,
        normalization=args.normalization,
        layernorm_epsilon=args.norm_epsilon,
        activation=args.activation,
        add_linear_bias=args.add_linear_bias,
        untie_embeddings_and_output_weights=args.untie_embeddings_and_output_weights,
        position_embedding_type=args.position_embedding_type,
        rotary_base=args.rotary_base,
        rotary_percent=args.rotary_percent,
        attention_dropout=args.attention_dropout,
        hidden_dropout=args.hidden_dropout,
        window_size=args.window_size,
        qk_layernorm=args.qk_layernorm,
        init_method_std=args.init_method_std,
        mtp_num_layers=args.mtp_num_layers,
        multi_latent_attention=args.multi_latent_attention,
        mamba_state_dim=args.mamba_state_dim,
        mamba_head_dim=args.mamba_head_dim,
        mamba_num_groups=args.mamba_num_groups,
        hybrid_override_pattern=args.hybrid_override_pattern,
        hybrid_attention_ratio=args.hybrid_attention_ratio,
        hybrid_mlp_ratio=args.hybrid_mlp_ratio,
        q_lora_rank=args.q_lora_rank,
        kv_lora_rank=args.kv_lora_rank,
        qk_nope_head_dim=args.qk_nope_head_dim,
        qk_rope_head_dim=args.qk_rope_head_dim,
        v_head_dim=args.v_head_dim,
        mtp_loss_scaling_factor=args.mtp_loss_scaling_factor,
        num_experts=args.num_experts,
        moe_router_topk=args.moe_router_topk,
        moe_ffn_hidden_size=args.moe_ffn_hidden_size,
        moe_aux_loss_coeff=args.moe_aux_loss_coeff,
        moe_z_loss_coeff=args.moe_z_loss_coeff,
        moe_router_score_function=args.moe_router_score_function,
        moe_shared_expert_intermediate_size=args.moe_shared_expert_intermediate_size,
        moe_token_dispatcher_type=args.moe_token_dispatcher_type,
        moe_layer_freq=args.moe_layer_freq,
        moe_router_force_load_balancing=args.moe_router_force_load_balancing,
        moe_router_pre_softmax=args.moe_router_pre_softmax,
        moe_router_num_groups=args.moe_router_num_groups,
        moe_router_group_topk=args.moe_router_group_topk,
        moe_input_jitter_eps=args.moe_input_jitter_eps,
        moe_aux_loss_type=args.moe_aux_loss_type,
        moe_router_enable_expert_bias=args.moe_router_enable_expert_bias,
        moe_router_bias_update_rate=args.moe_router_bias_update_rate,
        moe_expert_capacity_factor=args.moe_expert_capacity_factor,
        moe_token_drop_policy=args.moe_token_drop_policy,
        moe_pad_expert_input_to_capacity=args.moe_pad_expert_input_to_capacity,
        qk_clip_threshold=args.qk_clip_threshold,
        label_smoothing=args.label_smoothing,
        overlap_moe_expert_parallel_comm=args.overlap_moe_expert_parallel_comm,
        rope_scaling={"factor": 8.0, "low_freq_factor": 1.0, "high_freq_factor": 4.0,
                      "original_max_position_embeddings": 8192} if args.rope_scaling else None,
        tensor_parallel_size=args.tensor_model_parallel_size,
        pipeline_parallel_size=args.pipeline_model_parallel_size,
        virtual_pipeline_parallel_size=args.virtual_pipeline_model_parallel_size,
        context_parallel_size=args.context_parallel_size,
        cp_comm_type=args.cp_comm_type,
        num_layers_in_first_pipeline_stage=args.num_layers_in_first_pipeline_stage,
        num_layers_in_last_pipeline_stage=args.num_layers_in_last_pipeline_stage,
        expert_parallel_size=args.expert_model_parallel_size,
        expert_tensor_parallel_size=args.expert_tensor_parallel_size,
        sequence_parallel=args.sequence_parallel,
        tp_comm_overlap=args.tp_comm_overlap,
        bf16=args.bf16,
        fp16=args.fp16,
        fp8=args.fp8_format,
        fp8_amax_history_len=args.fp8_amax_history_len,
        fp8_margin=args.fp8_margin,
        recompute_granularity=args.recompute_granularity,
        recompute_num_layers=args.recompute_num_layers,
        activation_cpu_offload=args.activation_cpu_offload,
        recompute_modules=args.recompute_modules,
        activation_offload_layers=args.activation_offload_layers,
        deterministic_mode=args.deterministic_mode,
        gradient_accumulation_fusion=torch.cuda.is_available(),
    )
    opt_cfg = OptimizerConfig(
        optimizer=args.optimizer,
        sgd_momentum=args.sgd_momentum,
        optimizer_cpu_offload=args.optimizer_cpu_offload,
        muon_momentum=args.muon_momentum,
        lr=args.lr, min_lr=args.min_lr,
        decoupled_lr=args.decoupled_lr, decoupled_min_lr=args.decoupled_min_lr, weight_decay=args.weight_decay,
        adam_beta1=args.adam_beta1, adam_beta2=args.adam_beta2, adam_eps=args.adam_eps,
        clip_grad=args.clip_grad, bf16=args.bf16, fp16=args.fp16,
        loss_scale=args.loss_scale,
        use_distributed_optimizer=args.use_distributed_optimizer,
        overlap_param_gather=args.overlap_param_gather,
        lr_decay_style=args.lr_decay_style, lr_warmup_iters=args.lr_warmup_iters,
        lr_decay_iters=args.lr_decay_iters,
    )
    ddp_cfg = DDPConfig(
        grad_reduce_in_fp32=args.grad_reduce_in_fp32,
        overlap_grad_reduce=args.overlap_grad_reduce,
        use_distributed_optimizer=args.use_distributed_optimizer,
        bucket_size=args.bucket_size,
        check_for_nan_in_grad=args.check_for_nan_in_grad,
        num_distributed_optimizer_instances=args.num_distributed_optimizer_instances,
        use_rccl_registered_buffers=args.use_rccl_registered_buffers,
    )
    return cfg, opt_cfg, ddp_cfg


def apply_yaml_config(args, path: str, parser=None):
    """Overlay a yaml config file onto parsed args (reference
    training/yaml_arguments.py).  Keys use either underscores or the CLI's
    dashes; explicit CLI flags win over yaml values, yaml wins over
    defaults."""
    import yaml

    with open(path) as f:
        doc = yaml.safe_load(f) or {}
    if parser is None:
        parser = build_arg_parser()
    defaults = vars(parser.parse_args([]))
    unknown = []
    for key, value in doc.items():
        attr = key.replace("-", "_")
        if attr not in defaults:
            unknown.append(key)
            continue
        # only apply when the CLI left the default in place
        if getattr(args, attr) == defaults[attr]:
            setattr(args, attr, value)
    if unknown:
        raise ValueError(f"unknown yaml config keys: {unknown}")
    return args


def parse_and_validate_args(argv=None):
    import os

    parser = build_arg_parser()
    args = parser.parse_args(argv)
    if getattr(args, "yaml_cfg", None):
        apply_yaml_config(args, args.yaml_cfg, parser)
    import torch.distributed as _dist

    if _dist.is_initialized():
        args.world_size = _dist.get_world_size()
    else:
        args.world_size = int(os.environ.get("WORLD_SIZE", "1"))
    args.rank = int(os.environ.get("RANK", "0"))
    validate_args(args)
    return args
