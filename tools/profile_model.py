"""Profile N training steps with torch.profiler (rocTracer on ROCm).

    python tools/profile_model.py --steps 5 [model flags]

Prints the top ops by self time and (on GPU) device time, and optionally
exports a chrome trace (--trace-out trace.json, viewable in perfetto).
For per-kernel hardware counters use rocprofv3 around bench.py instead
(see profiles/README.md).
"""

from __future__ import annotations

import sys

import torch

sys.path.insert(0, ".")


def profile_steps(model, batches, steps: int = 3, trace_out: str = None,
                  row_limit: int = 15) -> str:
    """Runs fwd+bwd on `batches` under the profiler; returns the op table."""
    activities = [torch.profiler.ProfilerActivity.CPU]
    if torch.cuda.is_available():
        activities.append(torch.profiler.ProfilerActivity.CUDA)
    with torch.profiler.profile(activities=activities) as prof:
        for i in range(steps):
            b = batches[i % len(batches)]
            loss = model(b["tokens"], labels=b["labels"]).mean()
            loss.backward()
            model.zero_grad(set_to_none=True)
    if trace_out:
        prof.export_chrome_trace(trace_out)
    key = "self_cuda_time_total" if torch.cuda.is_available() else "self_cpu_time_total"
    return prof.key_averages().table(sort_by=key, row_limit=row_limit)


def main(argv=None):
    from megatron_amd.models.gpt import GPTModel
    from megatron_amd.parallel import grid as G
    from megatron_amd.training.arguments import build_arg_parser, configs_from_args

    parser = build_arg_parser()
    parser.add_argument("--steps", type=int, default=3)
    parser.add_argument("--trace-out", type=str, default=None)
    args = parser.parse_args(argv)
    args.world_size, args.rank = 1, 0
    cfg, _, _ = configs_from_args(args)
    G.initialize_model_parallel(world_size=1, rank=0)
    dev = "cuda" if torch.cuda.is_available() else "cpu"
    model = GPTModel(cfg).to(dev)
    if args.bf16:
        model = model.bfloat16()
    g = torch.Generator().manual_seed(args.seed)
    batches = []
    for _ in range(2):
        t = torch.randint(0, cfg.vocab_size, (args.micro_batch_size, args.seq_length + 1),
                          generator=g).to(dev)
        batches.append({"tokens": t[:, :-1], "labels": t[:, 1:]})
    print(profile_steps(model, batches, steps=args.steps, trace_out=args.trace_out))


if __name__ == "__main__":
    main()
