"""The pretraining application.

Capability analog of reference megatron/training/training.py:1015
(pretrain -> initialize -> setup_model_and_optimizer -> train loop with
logging / eval / checkpoint-and-exit) — MI355X-first, one process per GPU
over RCCL.
"""

from __future__ import annotations

import os
import signal
import sys
import time
from typing import Callable, Optional

import torch
import torch.distributed as dist

from megatron_amd.checkpoint import load_checkpoint, save_checkpoint
from megatron_amd.datasets.mock import MockGPTDataIterator
from megatron_amd.optimizer import OptimizerParamScheduler
from megatron_amd.parallel import grid as G
from megatron_amd.parallel.random import model_parallel_seed
from megatron_amd.training.arguments import configs_from_args, parse_and_validate_args
from megatron_amd.training.flops import MI355X_BF16_DENSE_PEAK_TFLOPS, num_floating_point_operations
from megatron_amd.training.training import setup_model_and_optimizer, train_step
from megatron_amd.utils.metrics import MetricsLogger, append_progress_log
from megatron_amd.utils.rerun_state_machine import (
    RerunDataIterator,
    get_rerun_state_machine,
    initialize_rerun_state_machine,
)
from megatron_amd.utils.straggler import EnergyMonitor, StragglerDetector
from megatron_amd.utils.timers import Timers


def _print_rank0(*a):
    if not dist.is_initialized() or dist.get_rank() == 0:
        print(*a, flush=True)


def initialize(args):
    if getattr(args, "nccl_communicator_config_path", None):
        # must load BEFORE groups are created so the options apply
        from megatron_amd.parallel.comm_config import load_comm_config

        load_comm_config(args.nccl_communicator_config_path)
    if torch.cuda.is_available():
        local_rank = int(os.environ.get("LOCAL_RANK", args.rank))
        torch.cuda.set_device(local_rank)
    if args.world_size > 1 or "RANK" in os.environ:
        G.init_distributed()
        G.initialize_model_parallel(
            tensor_parallel_size=args.tensor_model_parallel_size,
            pipeline_parallel_size=args.pipeline_model_parallel_size,
            context_parallel_size=args.context_parallel_size,
            expert_parallel_size=args.expert_model_parallel_size,
            expert_tensor_parallel_size=args.expert_tensor_parallel_size,
            virtual_pipeline_parallel_size=args.virtual_pipeline_model_parallel_size,
        )
    else:
        G.initialize_model_parallel(world_size=1, rank=0)
    if args.deterministic_mode:
        torch.use_deterministic_algorithms(True, warn_only=True)
    model_parallel_seed(args.seed)


def build_data_iterator(args, device, start_sample: int = 0, split: str = "train"):
    grid = G.get_grid()
    if args.mock_data or not args.data_path:
        # deterministic restart: fold the consumed-sample count into the seed
        it = MockGPTDataIterator(args.micro_batch_size, args.seq_length, args.vocab_size,
                                 seed=args.seed + (0 if split == "train" else 7919) + start_sample,
                                 device=device, dp_rank=grid.rank_in("dp_cp"))
        return iter(it)
    from megatron_amd.datasets.gpt_dataset import build_gpt_train_iterator

    return build_gpt_train_iterator(args, device, grid.rank_in("dp_cp"), grid.size("dp_cp"),
                                    start_sample=start_sample, split=split)


@torch.no_grad()
def evaluate(forward_step, args, chunks, cfg, device) -> float:
    """Mean validation loss over --eval-iters batches (reference
    training.py:4202 evaluate)."""
    for m in chunks:
        m.eval()
    it = build_data_iterator(args, str(device), split="valid")
    total = torch.zeros((), dtype=torch.float32)
    ntok = torch.zeros((), dtype=torch.float64)
    for _ in range(args.eval_iters):
        batch = next(it)
        out, loss_func = forward_step(iter([batch]), chunks[-1])
        loss_sum, n, _ = loss_func(out if out.dim() == 2 else out)
        total += loss_sum.detach().float().cpu()
        ntok += float(n)
    if dist.is_initialized():
        grid = G.get_grid()
        group = grid.group("dp_cp")
        if group is not None and dist.get_world_size(group) > 1:
            t = torch.stack([total.double(), ntok])
            dist.all_reduce(t, group=group)
            total, ntok = t[0].float(), t[1]
    for m in chunks:
        m.train()
    return float(total / max(float(ntok), 1.0))


class DistSignalHandler:
    """SIGTERM-driven save-and-exit (reference training/dist_signal_handler.py):
    the flag is all-reduced so every rank exits at the same iteration."""

    def __init__(self, sig=signal.SIGTERM):
        self.signal_received = False
        try:
            signal.signal(sig, self._handler)
        except ValueError:
            pass  # not the main thread (tests)

    def _handler(self, signum, frame):
        self.signal_received = True

    def should_exit(self) -> bool:
        flag = torch.tensor([1.0 if self.signal_received else 0.0])
        if dist.is_initialized():
            dist.all_reduce(flag, op=dist.ReduceOp.MAX)
        return bool(flag.item())


def pretrain(model_provider: Callable, argv=None, forward_step_builder=None):
    args = parse_and_validate_args(argv)
    initialize(args)
    cfg, opt_cfg, ddp_cfg = configs_from_args(args)
    device = torch.device("cuda", torch.cuda.current_device()) if torch.cuda.is_available() else torch.device("cpu")
    timers = Timers()

    if getattr(args, "use_fsdp", False):
        from megatron_amd.training.training import setup_fsdp_model_and_optimizer

        assert cfg.pipeline_parallel_size == 1 and cfg.tensor_parallel_size == 1, (
            "--use-fsdp composes with DP and EP (TP/PP use DDP + ZeRO-1)")
        chunks, optimizer = setup_fsdp_model_and_optimizer(model_provider, cfg, opt_cfg, device=device)
    else:
        chunks, optimizer = setup_model_and_optimizer(model_provider, cfg, opt_cfg, ddp_cfg, device=device)
    scheduler = OptimizerParamScheduler(optimizer, opt_cfg, args.train_iters)
    if getattr(args, "gpu_sniff_test", False):
        from megatron_amd.utils.gpu_health import gpu_sniff_test

        problems = gpu_sniff_test()
        if problems:
            raise RuntimeError("GPU health check failed: " + "; ".join(problems))
    if getattr(args, "profile_ranges", False):
        from megatron_amd.utils.annotations import enable_annotations

        enable_annotations(True)
    if getattr(args, "memory_snapshot_path", None) and torch.cuda.is_available():
        # reference training.py:2859 --memory-snapshot-path
        torch.cuda.memory._record_memory_history(max_entries=100_000)
    initialize_rerun_state_machine(args.rerun_mode)
    fault_injector = None
    if getattr(args, "fault_injection_type", None):
        from megatron_amd.utils.fault_injection import FaultInjector, FaultInjectorConfig

        fault_injector = FaultInjector(FaultInjectorConfig(
            enabled=True, fault_type=args.fault_injection_type,
            at_iteration=args.fault_injection_iteration,
            ranks=list(args.fault_injection_ranks)), rank=args.rank)
    moe_stats = None
    if args.num_experts:
        from megatron_amd.moe.moe_logging import MoEStatsTracker

        moe_stats = MoEStatsTracker(chunks[0] if isinstance(chunks, list) else chunks)
    straggler = StragglerDetector(enabled=args.log_straggler,
                                  control_port=getattr(args, 'straggler_ctrlr_port', None))
    energy = EnergyMonitor() if args.log_energy else None
    metrics = MetricsLogger(args.tensorboard_dir, rank=args.rank,
                            use_wandb=args.use_wandb, wandb_project=args.wandb_project)

    from megatron_amd.training.theoretical_memory import format_report, report

    cfg.seq_length = args.seq_length
    _print_rank0(format_report(report(cfg, args.micro_batch_size, args.num_microbatches,
                                      args.data_parallel_size,
                                      args.use_distributed_optimizer,
                                      args.recompute_granularity == "full")))

    iteration = 0
    start_time = time.time()
    sig_handler = DistSignalHandler() if args.exit_signal_handler else None
    if args.load:
        try:
            from megatron_amd.checkpoint.checkpointing import resolve_resume_source

            _root, _it = resolve_resume_source(args.load, args.non_persistent_ckpt_dir)
            load_optim = not (args.finetune or args.no_load_optim)
            iteration = load_checkpoint(_root or args.load, chunks, optimizer, scheduler,
                                        load_rng=not args.no_load_rng and not args.finetune,
                                        load_optim=load_optim)
            if args.finetune:
                iteration = 0  # new run: fresh schedule over the loaded weights
            _print_rank0(f"loaded checkpoint at iteration {iteration}")
        except FileNotFoundError:
            _print_rank0(f"no checkpoint found in {args.load}; starting fresh")

    n_chunks = len(chunks)
    # dump the resolved run configuration (reference core/config_logger.py)
    if args.rank == 0 and args.save:
        import dataclasses
        import json as _json
        import os as _os

        _os.makedirs(args.save, exist_ok=True)
        with open(_os.path.join(args.save, "run_config.json"), "w") as f:
            _json.dump({"args": {k: str(v) for k, v in sorted(vars(args).items())},
                        "transformer_config": {k: str(v) for k, v in
                                               sorted(dataclasses.asdict(cfg).items())}},
                       f, indent=1)

    from megatron_amd.training.microbatches import MicrobatchCalculator

    mb_calc = MicrobatchCalculator(args.global_batch_size, args.micro_batch_size,
                                   args.data_parallel_size,
                                   rampup=getattr(args, "rampup_batch_size", None))
    consumed = iteration * args.global_batch_size  # dataloader resume point
    data_iters = [RerunDataIterator(build_data_iterator(args, str(device), start_sample=consumed))
                  for _ in range(n_chunks)]

    def forward_step(data_iterator, model):
        batch = next(data_iterator)
        if cfg.context_parallel_size > 1:
            from megatron_amd.parallel.context_parallel import get_batch_on_this_cp_rank

            batch = get_batch_on_this_cp_rank(batch, mode=cfg.cp_comm_type)
        psp = None
        if getattr(args, "packed_sequences", False) or getattr(args, "reset_attention_mask", False):
            # THD training: flatten the microbatch into one packed stream;
            # per-row cu_seqlens (or one doc per row) shift to global offsets
            from megatron_amd.transformer.packed_seq import PackedSeqParams

            b, s = batch["tokens"].shape
            if "cu_seqlens" in batch:
                cu = [0]
                for r in range(b):
                    row_cu = [int(x) for x in batch["cu_seqlens"][r] if int(x) > 0]
                    base = r * s
                    cu.extend(base + x for x in row_cu if base + x > cu[-1])
                lengths = [b2 - a2 for a2, b2 in zip(cu, cu[1:])]
            else:
                lengths = [s] * b
            psp = PackedSeqParams.from_lengths(lengths, device=batch["tokens"].device)
            batch = dict(batch)
            batch["tokens"] = batch["tokens"].reshape(1, b * s)
            batch["labels"] = batch["labels"].reshape(1, b * s)
            if "loss_mask" in batch:
                batch["loss_mask"] = batch["loss_mask"].reshape(1, b * s)

        def loss_func(loss_sb):
            if "loss_mask" in batch:
                mask = batch["loss_mask"].transpose(0, 1)
                s = (loss_sb * mask).sum()
                ntok = mask.sum().long()
            else:
                s = loss_sb.sum()
                ntok = torch.tensor(loss_sb.numel(), device=loss_sb.device)
            # rerun machine: flag NaN/Inf losses for replay classification
            get_rerun_state_machine().validate_result(
                s, lambda t: not bool(torch.isfinite(t).all()), "nan/inf loss")
            return s, ntok, {"loss_sum": s.detach()}

        if psp is not None:
            out = model(batch["tokens"], labels=batch["labels"], packed_seq_params=psp)
        else:
            out = model(batch["tokens"], labels=batch["labels"])
        return out, loss_func

    if forward_step_builder is not None:
        forward_step = forward_step_builder(args)

    flops_per_iter = num_floating_point_operations(cfg, args.global_batch_size, args.seq_length)
    n_gpus = args.world_size
    _print_rank0(f"training: {args.train_iters} iters, GBS {args.global_batch_size}, "
                 f"microbatches {args.num_microbatches}, dp {args.data_parallel_size}")

    graphs_pending = bool(getattr(args, "hip_graphs", False)) and torch.cuda.is_available()

    def _capture_graphs():
        """--hip-graphs: per-layer fwd/bwd capture after the first iteration
        (a clean step boundary), under no_sync so capture-time grad-ready
        callbacks cannot launch reduces (see transformer/hip_graphs.py)."""
        import contextlib as _ctx

        from megatron_amd.transformer.hip_graphs import capture_block_hip_graphs

        core = chunks[0].module if hasattr(chunks[0], "module") else chunks[0]
        if args.num_experts or cfg.pipeline_parallel_size > 1 or n_chunks > 1:
            _print_rank0("--hip-graphs: skipped (MoE routing / PP chunks are dynamic)")
            return
        s_local = args.seq_length
        if cfg.sequence_parallel and cfg.tensor_parallel_size > 1:
            s_local //= cfg.tensor_parallel_size
        sample = torch.randn(s_local, args.micro_batch_size, cfg.hidden_size,
                             device=device, dtype=cfg.params_dtype)
        freqs = core._rotary_freqs(args.seq_length, device)
        with _ctx.ExitStack() as stack:
            for ch in chunks:
                if hasattr(ch, "no_sync"):
                    stack.enter_context(ch.no_sync())
            n = capture_block_hip_graphs(core.decoder, sample, rotary_freqs=freqs)
        _print_rank0(f"--hip-graphs: captured {n} layer graphs")

    prof = None
    while iteration < args.train_iters:
        if graphs_pending and iteration >= 1:
            _capture_graphs()
            graphs_pending = False
        if args.profile and iteration == args.profile_step_start and args.rank == 0:
            prof = torch.profiler.profile(
                activities=[torch.profiler.ProfilerActivity.CPU, torch.profiler.ProfilerActivity.CUDA],
                on_trace_ready=torch.profiler.tensorboard_trace_handler(args.profile_dir),
            )
            prof.__enter__()
        timers("iteration").start()
        straggler.start()
        cur_gbs, cur_nmb = mb_calc.get(consumed)
        result = train_step(forward_step, data_iters, chunks, optimizer, cfg,
                            cur_nmb, args.seq_length, args.micro_batch_size)
        consumed += cur_gbs
        straggler.stop()
        timers("iteration").stop()
        if fault_injector is not None:
            fault_injector.maybe_inject(iteration)
        if "exit_code" in result:
            if args.save:
                save_checkpoint(args.save, chunks, optimizer, iteration, scheduler)
            _print_rank0(f"rerun state machine requested exit (code {result['exit_code']})")
            metrics.close()
            sys.exit(result["exit_code"])
        iteration += 1
        scheduler.step()
        if moe_stats is not None:
            moe_stats.collect()
        if (args.non_persistent_save_interval and args.non_persistent_ckpt_dir
                and iteration % args.non_persistent_save_interval == 0):
            from megatron_amd.checkpoint.checkpointing import save_non_persistent_checkpoint

            save_non_persistent_checkpoint(args.non_persistent_ckpt_dir, chunks,
                                           optimizer, iteration, scheduler)
        if (args.check_weight_hash_across_dp_replicas_interval
                and iteration % args.check_weight_hash_across_dp_replicas_interval == 0):
            from megatron_amd.distributed.checks import check_param_hashes_across_dp_replicas

            if not check_param_hashes_across_dp_replicas(chunks):
                _print_rank0(f"WARNING: parameter hash mismatch across DP replicas at iteration {iteration}")
        if prof is not None and iteration == args.profile_step_end:
            prof.__exit__(None, None, None)
            prof = None

        if args.log_interval and iteration % args.log_interval == 0:
            t = timers("iteration").elapsed() / args.log_interval
            tokens_per_s = args.global_batch_size * args.seq_length / t
            tflops = flops_per_iter / t / max(n_gpus, 1) / 1e12
            msg = (f"iteration {iteration:6d}/{args.train_iters} | lm loss {result['lm_loss']:.4f} | "
                   f"lr {optimizer.get_lr():.3e} | iter time {t*1000:.1f}ms | tokens/s {tokens_per_s:.0f}")
            if args.log_throughput:
                msg += f" | TFLOP/s/GPU {tflops:.1f} | MFU {tflops/MI355X_BF16_DENSE_PEAK_TFLOPS*100:.1f}%"
            if moe_stats is not None:
                rep = moe_stats.report()
                worst = max((v["max_violation"] for v in rep.values()), default=0.0)
                msg += f" | moe max-violation {worst:.2f}"
            if result.get("grad_norm") is not None:
                msg += f" | grad norm {result['grad_norm']:.3f}"
            if args.log_params_norm:
                from megatron_amd.optimizer.clip import get_grad_norm, split_grads_for_norm

                ps = [p for c in chunks
                      for p in (c.module if hasattr(c, "module") else c).parameters()]
                dense, expert = split_grads_for_norm(ps, [p.data.float() for p in ps])
                msg += f" | params norm {float(get_grad_norm(dense, expert_grads=expert)):.3f}"
            if args.log_num_zeros_in_grad:
                nz = sum(int((getattr(p, "main_grad", None) if getattr(p, "main_grad", None)
                              is not None else (p.grad if p.grad is not None else p.new_zeros(1)))
                             .eq(0).sum())
                         for c in chunks
                         for p in (c.module if hasattr(c, "module") else c).parameters())
                msg += f" | zeros in grad {nz}"
            if args.log_memory and torch.cuda.is_available():
                msg += f" | mem {torch.cuda.max_memory_allocated()/2**30:.1f}GB"
            if energy is not None:
                e = energy.lap()
                if e is not None:
                    msg += f" | energy {e:.0f}J"
            _print_rank0(msg)
            metrics.log(iteration, lm_loss=result["lm_loss"], lr=optimizer.get_lr(),
                        iter_time_ms=t * 1000, tokens_per_s=tokens_per_s,
                        tflops_per_gpu=tflops, grad_norm=result.get("grad_norm"))
            if args.log_straggler and iteration % args.straggler_report_interval == 0:
                rep = straggler.report()
                if rep is not None:
                    _print_rank0(f"straggler: min rank {rep.min_rank} {rep.min_time_ms:.1f}ms | "
                                 f"max rank {rep.max_rank} {rep.max_time_ms:.1f}ms | "
                                 f"mean {rep.mean_time_ms:.1f}ms | power {rep.power_w}W | temp {rep.temp_c}C")

        if args.eval_interval and iteration % args.eval_interval == 0 and args.eval_iters:
            val = evaluate(forward_step, args, chunks, cfg, device)
            _print_rank0(f"validation loss at iteration {iteration}: {val:.4f}")
            metrics.log(iteration, valid_loss=val)

        if args.save and args.save_interval and iteration % args.save_interval == 0:
            save_checkpoint(args.save, chunks, optimizer, iteration, scheduler,
                            async_save=args.async_save)
            _print_rank0(f"saved checkpoint at iteration {iteration}")
        if args.exit_interval and iteration % args.exit_interval == 0:
            _print_rank0(f"exiting at iteration {iteration} (--exit-interval)")
            break
        if sig_handler is not None and sig_handler.should_exit():
            _print_rank0(f"SIGTERM received: checkpoint and exit at iteration {iteration}")
            if args.save:
                save_checkpoint(args.save, chunks, optimizer, iteration, scheduler)
            break
        if args.exit_duration_in_mins and (time.time() - start_time) / 60 > args.exit_duration_in_mins:
            _print_rank0(f"exiting after {args.exit_duration_in_mins} min (--exit-duration-in-mins)")
            if args.save:
                save_checkpoint(args.save, chunks, optimizer, iteration, scheduler)
            break

    if args.save:
        save_checkpoint(args.save, chunks, optimizer, iteration, scheduler)
        _print_rank0(f"saved final checkpoint at iteration {iteration}")
        append_progress_log(args.save, args.rank, f"finished at iteration {iteration}")
    if getattr(args, "memory_snapshot_path", None) and torch.cuda.is_available():
        torch.cuda.memory._dump_snapshot(args.memory_snapshot_path)
        torch.cuda.memory._record_memory_history(enabled=None)
    metrics.close()
    return iteration
