"""The pretraining application.

Capability analog of reference megatron/training/training.py:1015
(pretrain -> initialize -> setup_model_and_optimizer -> train loop with
logging / eval / checkpoint-and-exit) — MI355X-first, one process per GPU
over RCCL.
"""

from __future__ import annotations

import os
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
from megatron_amd.utils.timers import Timers


def _print_rank0(*a):
    if not dist.is_initialized() or dist.get_rank() == 0:
        print(*a, flush=True)


def initialize(args):
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


def build_data_iterator(args, device):
    grid = G.get_grid()
    if args.mock_data or not args.data_path:
        it = MockGPTDataIterator(args.micro_batch_size, args.seq_length, args.vocab_size,
                                 seed=args.seed, device=device, dp_rank=grid.rank_in("dp_cp"))
        return iter(it)
    from megatron_amd.datasets.gpt_dataset import build_gpt_train_iterator

    return build_gpt_train_iterator(args, device, grid.rank_in("dp_cp"), grid.size("dp_cp"))


def pretrain(model_provider: Callable, argv=None, forward_step_builder=None):
    args = parse_and_validate_args(argv)
    initialize(args)
    cfg, opt_cfg, ddp_cfg = configs_from_args(args)
    device = torch.device("cuda", torch.cuda.current_device()) if torch.cuda.is_available() else torch.device("cpu")
    timers = Timers()

    chunks, optimizer = setup_model_and_optimizer(model_provider, cfg, opt_cfg, ddp_cfg, device=device)
    scheduler = OptimizerParamScheduler(optimizer, opt_cfg, args.train_iters)

    iteration = 0
    if args.load:
        try:
            iteration = load_checkpoint(args.load, chunks, optimizer, scheduler,
                                        load_rng=not args.no_load_rng)
            _print_rank0(f"loaded checkpoint at iteration {iteration}")
        except FileNotFoundError:
            _print_rank0(f"no checkpoint found in {args.load}; starting fresh")

    n_chunks = len(chunks)
    data_iters = [build_data_iterator(args, str(device)) for _ in range(n_chunks)]

    def forward_step(data_iterator, model):
        batch = next(data_iterator)
        if cfg.context_parallel_size > 1:
            from megatron_amd.parallel.context_parallel import get_batch_on_this_cp_rank

            batch = get_batch_on_this_cp_rank(batch, mode=cfg.cp_comm_type)

        def loss_func(loss_sb):
            if "loss_mask" in batch:
                mask = batch["loss_mask"].transpose(0, 1)
                s = (loss_sb * mask).sum()
                ntok = mask.sum().long()
            else:
                s = loss_sb.sum()
                ntok = torch.tensor(loss_sb.numel(), device=loss_sb.device)
            return s, ntok, {"loss_sum": s.detach()}

        out = model(batch["tokens"], labels=batch["labels"])
        return out, loss_func

    if forward_step_builder is not None:
        forward_step = forward_step_builder(args)

    flops_per_iter = num_floating_point_operations(cfg, args.global_batch_size, args.seq_length)
    n_gpus = args.world_size
    _print_rank0(f"training: {args.train_iters} iters, GBS {args.global_batch_size}, "
                 f"microbatches {args.num_microbatches}, dp {args.data_parallel_size}")

    prof = None
    while iteration < args.train_iters:
        if args.profile and iteration == args.profile_step_start and args.rank == 0:
            prof = torch.profiler.profile(
                activities=[torch.profiler.ProfilerActivity.CPU, torch.profiler.ProfilerActivity.CUDA],
                on_trace_ready=torch.profiler.tensorboard_trace_handler(args.profile_dir),
            )
            prof.__enter__()
        timers("iteration").start()
        result = train_step(forward_step, data_iters, chunks, optimizer, cfg,
                            args.num_microbatches, args.seq_length, args.micro_batch_size)
        timers("iteration").stop()
        iteration += 1
        scheduler.step()
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
            if result.get("grad_norm") is not None:
                msg += f" | grad norm {result['grad_norm']:.3f}"
            if args.log_memory and torch.cuda.is_available():
                msg += f" | mem {torch.cuda.max_memory_allocated()/2**30:.1f}GB"
            _print_rank0(msg)

        if args.save and args.save_interval and iteration % args.save_interval == 0:
            save_checkpoint(args.save, chunks, optimizer, iteration, scheduler,
                            async_save=args.async_save)
            _print_rank0(f"saved checkpoint at iteration {iteration}")
        if args.exit_interval and iteration % args.exit_interval == 0:
            _print_rank0(f"exiting at iteration {iteration} (--exit-interval)")
            break

    if args.save:
        save_checkpoint(args.save, chunks, optimizer, iteration, scheduler)
        _print_rank0(f"saved final checkpoint at iteration {iteration}")
    return iteration
