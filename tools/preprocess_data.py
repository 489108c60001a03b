"""Corpus preprocessing: JSONL -> indexed (.bin/.idx) dataset.

Capability analog of reference tools/preprocess_data.py: tokenize a json-lines
corpus in parallel worker processes, append EOD per document, write the
MMIDIDX-compatible indexed dataset.

    python tools/preprocess_data.py --input corpus.jsonl --json-key text \
        --tokenizer-type HuggingFace --tokenizer-model <path> \
        --output-prefix /data/mycorpus --append-eod --workers 8
"""

from __future__ import annotations

import argparse
import json
import multiprocessing as mp
import sys
import time

import numpy as np

sys.path.insert(0, ".")  # allow running from the repo root

from megatron_amd.datasets.indexed import IndexedDatasetBuilder, optimal_token_dtype  # noqa: E402
from megatron_amd.tokenizers import build_tokenizer  # noqa: E402

_worker_tok = None
_worker_args = None


def _init_worker(args):
    global _worker_tok, _worker_args
    _worker_args = args
    _worker_tok = build_tokenizer(args.tokenizer_type, args.tokenizer_model, args.vocab_size)


def _encode(line: str):
    line = line.strip()
    if not line:
        return None
    doc = json.loads(line)
    text = doc[_worker_args.json_key]
    ids = _worker_tok.tokenize(text)
    if _worker_args.append_eod:
        ids = list(ids) + [_worker_tok.eod]
    return ids if ids else None


def main(argv=None):
    p = argparse.ArgumentParser()
    p.add_argument("--input", required=True, help="jsonl file")
    p.add_argument("--json-key", default="text")
    p.add_argument("--output-prefix", required=True)
    p.add_argument("--tokenizer-type", default="NullTokenizer")
    p.add_argument("--tokenizer-model", default=None)
    p.add_argument("--vocab-size", type=int, default=None)
    p.add_argument("--append-eod", action="store_true")
    p.add_argument("--workers", type=int, default=1)
    p.add_argument("--log-interval", type=int, default=10000)
    args = p.parse_args(argv)

    tok = build_tokenizer(args.tokenizer_type, args.tokenizer_model, args.vocab_size)
    dtype = optimal_token_dtype(tok.vocab_size)
    builder = IndexedDatasetBuilder(args.output_prefix, dtype=dtype)

    t0 = time.time()
    n_docs = n_tokens = 0
    with open(args.input, "r", encoding="utf-8") as f:
        if args.workers > 1:
            pool = mp.Pool(args.workers, initializer=_init_worker, initargs=(args,))
            stream = pool.imap(_encode, f, chunksize=32)
        else:
            _init_worker(args)
            stream = map(_encode, f)
        for ids in stream:
            if ids is None:
                continue
            builder.add_document(np.asarray(ids, dtype=dtype))
            n_docs += 1
            n_tokens += len(ids)
            if args.log_interval and n_docs % args.log_interval == 0:
                dt = time.time() - t0
                print(f"{n_docs} docs, {n_tokens} tokens, {n_docs/dt:.0f} docs/s", flush=True)
        if args.workers > 1:
            pool.close()
            pool.join()
    builder.finalize()
    print(f"wrote {args.output_prefix}.bin/.idx: {n_docs} docs, {n_tokens} tokens "
          f"({time.time()-t0:.1f}s)", flush=True)


if __name__ == "__main__":
    main()
