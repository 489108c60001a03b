"""Serve a model over REST (reference tools/run_text_generation_server.py).

    python tools/run_text_generation_server.py --num-layers ... --load <ckpt> \
        --tokenizer-type HuggingFace --tokenizer-model <path> --port 5000
"""

from __future__ import annotations

import sys

sys.path.insert(0, ".")

import torch  # noqa: E402

from megatron_amd.inference.engine import DynamicInferenceEngine  # noqa: E402
from megatron_amd.inference.server import run_server  # noqa: E402
from megatron_amd.models.gpt import GPTModel  # noqa: E402
from megatron_amd.tokenizers import build_tokenizer  # noqa: E402
from megatron_amd.training.arguments import build_arg_parser, configs_from_args  # noqa: E402
from megatron_amd.training.pretrain import initialize  # noqa: E402


def main(argv=None):
    parser = build_arg_parser()
    parser.add_argument("--port", type=int, default=5000)
    parser.add_argument("--host", type=str, default="127.0.0.1")
    parser.add_argument("--kv-blocks", type=int, default=4096)
    parser.add_argument("--kv-block-size", type=int, default=256)
    parser.add_argument("--kv-cache-dtype", choices=["fp8"], default=None,
                        help="fp8: e4m3 KV payload + per-slot scales (half the cache memory)")
    parser.add_argument("--disable-prefix-caching", action="store_true")
    args = parser.parse_args(argv)
    import os

    args.world_size = int(os.environ.get("WORLD_SIZE", "1"))
    args.rank = int(os.environ.get("RANK", "0"))
    from megatron_amd.training.arguments import validate_args

    validate_args(args)
    initialize(args)
    cfg, _, _ = configs_from_args(args)
    device = torch.device("cuda") if torch.cuda.is_available() else torch.device("cpu")
    model = GPTModel(cfg).to(device)
    if args.load:
        from megatron_amd.checkpoint import load_checkpoint

        load_checkpoint(args.load, [model], None, None, load_rng=False)
    tokenizer = build_tokenizer(args.tokenizer_type, args.tokenizer_model, args.vocab_size)
    engine = DynamicInferenceEngine(model, tokenizer, num_blocks=args.kv_blocks,
                                    block_size=args.kv_block_size, device=device,
                                    kv_cache_dtype=args.kv_cache_dtype,
                                    enable_prefix_caching=not args.disable_prefix_caching)
    print(f"serving on {args.host}:{args.port}", flush=True)
    run_server(engine, tokenizer, host=args.host, port=args.port)


if __name__ == "__main__":
    main()
