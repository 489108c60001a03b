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
    raise ValueError(f"unknown tokenizer type {tokenizer_type}")


def pad_vocab_size(vocab_size: int, multiple: int, tp_size: int) -> int:
    """Pad to a multiple of (multiple * tp) for clean TP sharding
    (reference training/tokenizer _vocab_size_with_padding)."""
    m = multiple * tp_size
    return ((vocab_size + m - 1) // m) * m


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
