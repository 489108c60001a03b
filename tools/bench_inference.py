"""Serving throughput micro-benchmark (analog of the reference's
tests/performance_tests inference baselines: synthetic ISL/OSL, batch decode).

    python tools/bench_inference.py --batch 128 --isl 512 --osl 128
"""

from __future__ import annotations

import argparse
import json
import sys
import time

sys.path.insert(0, ".")

import torch  # noqa: E402

from megatron_amd.config import TransformerConfig  # noqa: E402
from megatron_amd.inference import DynamicInferenceEngine, SamplingParams  # noqa: E402
from megatron_amd.models.gpt import GPTModel  # noqa: E402
from megatron_amd.parallel import grid as G  # noqa: E402
from megatron_amd.parallel.random import model_parallel_seed  # noqa: E402


def main(argv=None):
    p = argparse.ArgumentParser()
    p.add_argument("--layers", type=int, default=24)
    p.add_argument("--hidden", type=int, default=1024)
    p.add_argument("--heads", type=int, default=16)
    p.add_argument("--kv-groups", type=int, default=8)
    p.add_argument("--ffn", type=int, default=4096)
    p.add_argument("--vocab", type=int, default=32768)
    p.add_argument("--batch", type=int, default=128)
    p.add_argument("--isl", type=int, default=512)
    p.add_argument("--osl", type=int, default=128)
    p.add_argument("--block-size", type=int, default=256)
    p.add_argument("--kv-cache-dtype", choices=["fp8"], default=None)
    p.add_argument("--speculative", action="store_true",
                   help="static-engine draft-verify decode (prompt-lookup drafts)")
    args = p.parse_args(argv)

    assert torch.cuda.is_available()
    G.initialize_model_parallel(world_size=1, rank=0)
    model_parallel_seed(1234)
    cfg = TransformerConfig(
        num_layers=args.layers, hidden_size=args.hidden, num_attention_heads=args.heads,
        num_query_groups=args.kv_groups, ffn_hidden_size=args.ffn, vocab_size=args.vocab,
        max_position_embeddings=args.isl + args.osl + 8, bf16=True)
    with torch.device("cuda"):
        model = GPTModel(cfg).eval()
    n_params = sum(x.numel() for x in model.parameters())

    max_tokens_total = args.batch * (args.isl + args.osl + args.block_size)
    num_blocks = (max_tokens_total + args.block_size - 1) // args.block_size
    eng = DynamicInferenceEngine(model, num_blocks=num_blocks, block_size=args.block_size,
                                 max_batch=args.batch, kv_cache_dtype=args.kv_cache_dtype)
    rng = torch.Generator().manual_seed(0)
    prompts = [torch.randint(0, args.vocab, (args.isl,), generator=rng).tolist()
               for _ in range(args.batch)]
    params = SamplingParams(max_tokens=args.osl, greedy=True, stop_on_eod=False)

    if args.speculative:
        from megatron_amd.inference import StaticInferenceEngine

        eng = StaticInferenceEngine(model, max_batch=args.batch,
                                    max_seq=args.isl + args.osl + 16)
        run = lambda ps, pr: eng.generate_speculative(ps, pr, num_draft=3)
    else:
        run = eng.generate
    # warmup: one tiny round
    run(prompts[:2], SamplingParams(max_tokens=4, greedy=True, stop_on_eod=False))
    torch.cuda.synchronize()
    t0 = time.time()
    results = run(prompts, params)
    torch.cuda.synchronize()
    dt = time.time() - t0
    out_tokens = sum(len(r.output_tokens) for r in results)
    print(json.dumps({
        "metric": "serving_tokens_per_s", "value": round(out_tokens / dt, 1),
        "unit": "output tok/s", "tpot_ms": round(1000 * dt / args.osl, 2),
        "elapsed_s": round(dt, 2), "batch": args.batch, "isl": args.isl, "osl": args.osl,
        "params_m": round(n_params / 1e6, 1), "dtype": "bf16", "data": "synthetic",
        "kv_cache_dtype": args.kv_cache_dtype or "bf16", "speculative": args.speculative,
    }))


if __name__ == "__main__":
    main()
