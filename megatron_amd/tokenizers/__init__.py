"""Tokenizer library (capability analog of reference megatron/core/tokenizers/:
build_tokenizer + sentencepiece / HuggingFace / tiktoken / byte-level / null
libraries). Uniform interface: tokenize/detokenize, vocab_size, eod."""

from __future__ import annotations

from typing import List, Optional


class MegatronTokenizer:
    """Abstract tokenizer interface."""

    def tokenize(self, text: str) -> List[int]:
        raise NotImplementedError

    def detokenize(self, ids: List[int]) -> str:
        raise NotImplementedError

    @property
    def vocab_size(self) -> int:
        raise NotImplementedError

    @property
    def eod(self) -> int:
        raise NotImplementedError

    @property
    def pad(self) -> int:
        return self.eod


class NullTokenizer(MegatronTokenizer):
    """Space-separated integer ids; for mock/synthetic pipelines
    (reference tokenizers NullTokenizer)."""

    def __init__(self, vocab_size: int = 131072):
        self._vocab_size = int(vocab_size)

    def tokenize(self, text: str) -> List[int]:
        return [int(t) for t in text.split()]

    def detokenize(self, ids) -> str:
        return " ".join(str(int(i)) for i in ids)

    @property
    def vocab_size(self) -> int:
        return self._vocab_size

    @property
    def eod(self) -> int:
        return self._vocab_size - 1


class ByteLevelTokenizer(MegatronTokenizer):
    """UTF-8 bytes + specials (reference byte-level library)."""

    EOD = 256

    def tokenize(self, text: str) -> List[int]:
        return list(text.encode("utf-8"))

    def detokenize(self, ids) -> str:
        return bytes(i for i in ids if i < 256).decode("utf-8", errors="replace")

    @property
    def vocab_size(self) -> int:
        return 257

    @property
    def eod(self) -> int:
        return self.EOD


class HuggingFaceTokenizer(MegatronTokenizer):
    def __init__(self, model_name_or_path: str):
        from transformers import AutoTokenizer

        self._tok = AutoTokenizer.from_pretrained(model_name_or_path)

    def tokenize(self, text: str) -> List[int]:
        return self._tok.encode(text, add_special_tokens=False)

    def detokenize(self, ids) -> str:
        return self._tok.decode(ids)

    def apply_chat_template(self, messages, add_generation_prompt: bool = True) -> List[int]:
        """Chat-format token ids via the model's HF chat template
        (serving-side convenience; messages = [{'role', 'content'}, ...])."""
        return self._tok.apply_chat_template(
            messages, add_generation_prompt=add_generation_prompt, tokenize=True)

    @property
    def vocab_size(self) -> int:
        return len(self._tok)

    @property
    def eod(self) -> int:
        t = self._tok
        return t.eos_token_id if t.eos_token_id is not None else t.pad_token_id


class SentencePieceTokenizer(MegatronTokenizer):
    def __init__(self, model_file: str):
        import sentencepiece as spm

        self._sp = spm.SentencePieceProcessor(model_file=model_file)

    def tokenize(self, text: str) -> List[int]:
        return self._sp.encode(text)

    def detokenize(self, ids) -> str:
        return self._sp.decode(list(int(i) for i in ids))

    @property
    def vocab_size(self) -> int:
        return self._sp.get_piece_size()

    @property
    def eod(self) -> int:
        eos = self._sp.eos_id()
        return eos if eos >= 0 else self._sp.get_piece_size() - 1


class TiktokenTokenizer(MegatronTokenizer):
    def __init__(self, encoding_name: str = "cl100k_base"):
        try:
            import tiktoken
        except ImportError as e:
            raise ImportError("tiktoken is not installed in this image") from e
        self._enc = tiktoken.get_encoding(encoding_name)

    def tokenize(self, text: str) -> List[int]:
        return self._enc.encode(text)

    def detokenize(self, ids) -> str:
        return self._enc.decode(list(ids))

    @property
    def vocab_size(self) -> int:
        return self._enc.n_vocab

    @property
    def eod(self) -> int:
        return self._enc.eot_token


class GPT2BPETokenizer(MegatronTokenizer):
    """Self-contained byte-level BPE over local vocab.json + merges.txt
    (reference megatron/training/tokenizer GPT2BPETokenizer role): no
    network, no HF dependency.  Byte-level means any input round-trips:
    raw bytes map to 256 printable unicode marks, BPE merges apply on top."""

    def __init__(self, vocab_file: str, merges_file: str,
                 special_tokens: Optional[List[str]] = None):
        import json

        import regex

        with open(vocab_file, encoding="utf-8") as f:
            self.encoder = json.load(f)
        self.decoder = {v: k for k, v in self.encoder.items()}
        with open(merges_file, encoding="utf-8") as f:
            merges = [tuple(line.split()) for line in f.read().split("\n")
                      if line and not line.startswith("#") and len(line.split()) == 2]
        self.bpe_ranks = {m: i for i, m in enumerate(merges)}
        self.byte_encoder = self._bytes_to_unicode()
        self.byte_decoder = {c: b for b, c in self.byte_encoder.items()}
        # the GPT-2 pre-tokenization pattern (contractions, letter/number
        # runs, punctuation runs, whitespace)
        self.pat = regex.compile(
            r"'s|'t|'re|'ve|'m|'ll|'d| ?\p{L}+| ?\p{N}+| ?[^\s\p{L}\p{N}]+|\s+(?!\S)|\s+")
        self._cache: dict = {}
        self._eod = self.encoder.get("<|endoftext|>", len(self.encoder) - 1)
        for tok in special_tokens or []:
            if tok not in self.encoder:
                self.encoder[tok] = len(self.encoder)
                self.decoder[self.encoder[tok]] = tok

    @staticmethod
    def _bytes_to_unicode():
        # printable ranges map to themselves; everything else is shifted into
        # the 256+ private block so merges files stay printable
        bs = (list(range(ord("!"), ord("~") + 1)) + list(range(0xA1, 0xAD))
              + list(range(0xAE, 0x100)))
        cs = bs[:]
        n = 0
        for b in range(256):
            if b not in bs:
                bs.append(b)
                cs.append(256 + n)
                n += 1
        return dict(zip(bs, [chr(c) for c in cs]))

    def _bpe(self, token: str) -> List[str]:
        if token in self._cache:
            return self._cache[token]
        word = list(token)
        while len(word) > 1:
            pairs = {(word[i], word[i + 1]) for i in range(len(word) - 1)}
            best = min(pairs, key=lambda p: self.bpe_ranks.get(p, float("inf")))
            if best not in self.bpe_ranks:
                break
            a, b = best
            out = []
            i = 0
            while i < len(word):
                if i < len(word) - 1 and word[i] == a and word[i + 1] == b:
                    out.append(a + b)
                    i += 2
                else:
                    out.append(word[i])
                    i += 1
            word = out
        self._cache[token] = word
        return word

    def tokenize(self, text: str) -> List[int]:
        ids: List[int] = []
        for chunk in self.pat.findall(text):
            mapped = "".join(self.byte_encoder[b] for b in chunk.encode("utf-8"))
            ids.extend(self.encoder[p] for p in self._bpe(mapped))
        return ids

    def detokenize(self, ids) -> str:
        text = "".join(self.decoder[int(i)] for i in ids)
        raw = bytes(self.byte_decoder[c] for c in text)
        return raw.decode("utf-8", errors="replace")

    @property
    def vocab_size(self) -> int:
        return len(self.encoder)

    @property
    def eod(self) -> int:
        return self._eod


def build_tokenizer(tokenizer_type: str, tokenizer_model: Optional[str] = None,
                    vocab_size: Optional[int] = None) -> MegatronTokenizer:
    t = tokenizer_type.lower()
    if t in ("nulltokenizer", "null"):
        return NullTokenizer(vocab_size or 131072)
    if t in ("bytelevel", "byte-level"):
        return ByteLevelTokenizer()
    if t in ("huggingfacetokenizer", "huggingface", "hf"):
        assert tokenizer_model, "--tokenizer-model required for HuggingFace"
        return HuggingFaceTokenizer(tokenizer_model)
    if t in ("sentencepiecetokenizer", "sentencepiece", "spm"):
        assert tokenizer_model, "--tokenizer-model required for SentencePiece"
        return SentencePieceTokenizer(tokenizer_model)
    if t in ("tiktokentokenizer", "tiktoken"):
        return TiktokenTokenizer(tokenizer_model or "cl100k_base")
    if t in ("gpt2bpetokenizer", "gpt2", "gpt2-bpe"):
        assert tokenizer_model, "--tokenizer-model VOCAB,MERGES required for GPT2 BPE"
        vocab, merges = tokenizer_model.split(",")
        return GPT2BPETokenizer(vocab, merges)
    raise ValueError(f"unknown tokenizer type {tokenizer_type}")


# ---------------------------------------------------------------------------
# serving / training utilities (reference tokenizers/ chat templates,
# vocab padding, and the streaming detokenizer the SSE server needs)
# ---------------------------------------------------------------------------


def pad_vocab_size(vocab_size: int, tensor_parallel_size: int = 1,
                   make_divisible_by: int = 128) -> int:
    """Pad the vocabulary so each TP shard is a multiple of
    ``make_divisible_by`` (reference _vocab_size_with_padding): keeps the
    output-layer GEMM and vocab-parallel CE on aligned shard sizes."""
    mult = make_divisible_by * tensor_parallel_size
    return ((vocab_size + mult - 1) // mult) * mult


class IncrementalDetokenizer:
    """Streaming detokenization for SSE serving: UTF-8 multi-byte sequences
    (and tokenizers whose pieces merge across boundaries) cannot be decoded
    token-by-token; this holds back the undecodable tail and emits only the
    stable prefix delta per step (reference inference text-gen controller
    detokenize-incrementally behavior)."""

    REPLACEMENT = "�"

    def __init__(self, tokenizer: MegatronTokenizer):
        self._tok = tokenizer
        self._ids: List[int] = []
        self._emitted = 0  # chars already streamed out

    def put(self, token_id: int) -> str:
        """Add one token; returns the newly stable text delta ('' if the
        tail is still an incomplete sequence)."""
        self._ids.append(int(token_id))
        text = self._tok.detokenize(self._ids)
        # hold back a trailing replacement char (incomplete utf-8 etc.)
        stable_end = len(text)
        while stable_end > 0 and text[stable_end - 1] == self.REPLACEMENT:
            stable_end -= 1
        delta = text[self._emitted:stable_end]
        self._emitted = stable_end
        return delta

    def flush(self) -> str:
        """Final flush: emit whatever remains (incl. replacement chars)."""
        text = self._tok.detokenize(self._ids)
        delta = text[self._emitted:]
        self._emitted = len(text)
        return delta


def apply_chat_template(tokenizer: MegatronTokenizer, messages: List[dict],
                        add_generation_prompt: bool = True) -> List[int]:
    """Render a chat conversation to token ids.

    HF tokenizers with a built-in chat template use it; every other
    tokenizer gets a simple generic template (role-tagged lines), so the
    /v1/chat endpoint works with any backend (reference tokenizers/text
    model-specific parsers)."""
    hf = getattr(tokenizer, "_tok", None)
    if hf is not None and getattr(hf, "chat_template", None):
        return list(hf.apply_chat_template(
            messages, add_generation_prompt=add_generation_prompt, tokenize=True))
    parts = []
    for m in messages:
        parts.append(f"<|{m.get('role', 'user')}|>\n{m.get('content', '')}\n")
    if add_generation_prompt:
        parts.append("<|assistant|>\n")
    return tokenizer.tokenize("".join(parts))
