"""Offline evaluation: perplexity and cloze (last-token) accuracy.

Capability analog of the reference's tasks/ zero-shot evaluation
(main.py --task LAMBADA / WIKITEXT103: tasks/zeroshot_gpt/evaluate.py):

    # perplexity over the validation split of an indexed corpus
    python tools/evaluate.py ppl --load <ckpt> --data-path <prefix> [model flags]

    # cloze accuracy: each line of --cloze-file is a JSON list of token ids;
    # the model must predict the final token from the preceding context
    python tools/evaluate.py cloze --load <ckpt> --cloze-file <jsonl> [model flags]

Importable API (used by unit tests): `evaluate_perplexity(model, batches)`
and `evaluate_cloze(model, samples)`.
"""

from __future__ import annotations

import json
import math
import sys
from typing import Iterable, List, Sequence

import torch

sys.path.insert(0, ".")


@torch.no_grad()
def evaluate_perplexity(model, batches: Iterable[dict]) -> dict:
    """batches: dicts with tokens [b, s] and labels [b, s] (+optional
    loss_mask).  Returns {'ppl', 'loss', 'tokens'}."""
    total_loss, total_tokens = 0.0, 0
    for batch in batches:
        loss_sb = model(batch["tokens"], labels=batch["labels"])  # [s, b]
        mask = batch.get("loss_mask")
        if mask is not None:
            m = mask.transpose(0, 1).to(loss_sb.dtype)
            total_loss += float((loss_sb * m).sum())
            total_tokens += int(m.sum())
        else:
            total_loss += float(loss_sb.sum())
            total_tokens += loss_sb.numel()
    mean = total_loss / max(total_tokens, 1)
    return {"ppl": math.exp(min(mean, 20.0)), "loss": mean, "tokens": total_tokens}


@torch.no_grad()
def evaluate_cloze(model, samples: Sequence[Sequence[int]]) -> dict:
    """LAMBADA-style: predict the last token of each sample greedily from
    the preceding context.  Returns {'accuracy', 'correct', 'total'}."""
    correct = 0
    for toks in samples:
        ctx, target = list(toks[:-1]), int(toks[-1])
        logits = model(torch.tensor([ctx]))  # [s, 1, V]
        pred = int(logits[-1, 0].float().argmax())
        correct += int(pred == target)
    total = len(samples)
    return {"accuracy": correct / max(total, 1), "correct": correct, "total": total}


def main(argv=None):
    from megatron_amd.checkpoint.checkpointing import load_checkpoint
    from megatron_amd.models.gpt import GPTModel
    from megatron_amd.parallel import grid as G
    from megatron_amd.training.arguments import build_arg_parser, configs_from_args

    parser = build_arg_parser()
    parser.add_argument("mode", choices=["ppl", "cloze"])
    parser.add_argument("--cloze-file", type=str, default=None)
    parser.add_argument("--eval-samples", type=int, default=64)
    args = parser.parse_args(argv)
    args.world_size, args.rank = 1, 0
    cfg, _, _ = configs_from_args(args)
    G.initialize_model_parallel()
    model = GPTModel(cfg).eval()
    if args.load:
        load_checkpoint(args.load, [model], None, load_rng=False)

    if args.mode == "cloze":
        samples = [json.loads(l) for l in open(args.cloze_file)]
        print(json.dumps(evaluate_cloze(model, samples)))
        return
    # ppl over the validation split
    from megatron_amd.datasets.gpt_dataset import build_gpt_datasets

    _, valid, _ = build_gpt_datasets(args.data_path, args.seq_length,
                                     seed=args.seed, train_samples=args.eval_samples,
                                     split=args.split)
    batches = []
    for i in range(min(args.eval_samples, len(valid))):
        s = valid[i]
        batches.append({"tokens": s["tokens"].unsqueeze(0), "labels": s["labels"].unsqueeze(0)})
    print(json.dumps(evaluate_perplexity(model, batches)))


if __name__ == "__main__":
    main()
