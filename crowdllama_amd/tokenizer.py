"""Tokenizer fed from the GGUF vocab section (reference parity: the C++
tokenizer inside llama.cpp that Ollama uses — SURVEY.md §2.3 row
"Tokenizer").

Two implementations with identical behavior:
- Tokenizer: pure-Python reference (CPU tests, fallback),
- NativeTokenizer: the C++ implementation in the engine extension
  (ops/csrc/tokenizer.cpp), used by HipEngine.

Supports byte-fallback vocabularies (<0xNN> byte tokens, as the synthetic
checkpoints emit) and GPT-2-style byte-level BPE when
`tokenizer.ggml.merges` is present.
"""

from __future__ import annotations

import functools


@functools.lru_cache(maxsize=1)
def _bytes_to_unicode() -> dict[int, str]:
    """GPT-2 byte<->unicode table (standard public construction)."""
    bs = (list(range(ord("!"), ord("~") + 1))
          + list(range(0xA1, 0xAD)) + list(range(0xAE, 0x100)))
    cs = bs[:]
    n = 0
    for b in range(256):
        if b not in bs:
            bs.append(b)
            cs.append(256 + n)
            n += 1
    return dict(zip(bs, map(chr, cs)))


class Tokenizer:
    def __init__(self, tokens: list[str], merges: list[str] | None = None,
                 bos_id: int = 1, eos_id: int = 2, model: str = "gpt2"):
        self.tokens = tokens
        self.vocab = {t: i for i, t in enumerate(tokens)}
        self.bos_id = bos_id
        self.eos_id = eos_id
        self.model = model
        self.byte_tokens: dict[int, int] = {}
        for i, t in enumerate(tokens):
            if len(t) == 6 and t.startswith("<0x") and t.endswith(">"):
                try:
                    self.byte_tokens[int(t[3:5], 16)] = i
                except ValueError:
                    pass
        self.merge_ranks: dict[tuple[str, str], int] = {}
        if merges:
            for rank, m in enumerate(merges):
                a, _, b = m.partition(" ")
                self.merge_ranks[(a, b)] = rank

    @classmethod
    def from_gguf(cls, reader) -> "Tokenizer":
        md = reader.metadata
        return cls(
            tokens=md.get("tokenizer.ggml.tokens", []),
            merges=md.get("tokenizer.ggml.merges"),
            bos_id=md.get("tokenizer.ggml.bos_token_id", 1),
            eos_id=md.get("tokenizer.ggml.eos_token_id", 2),
            model=md.get("tokenizer.ggml.model", "gpt2"),
        )

    # ------------------------------------------------------------- encode

    def _bpe(self, word: str) -> list[str]:
        parts = list(word)
        while len(parts) > 1:
            best, best_rank = None, 1 << 60
            for i in range(len(parts) - 1):
                r = self.merge_ranks.get((parts[i], parts[i + 1]))
                if r is not None and r < best_rank:
                    best, best_rank = i, r
            if best is None:
                break
            parts[best:best + 2] = [parts[best] + parts[best + 1]]
        return parts

    def encode(self, text: str, add_bos: bool = True) -> list[int]:
        ids: list[int] = []
        if add_bos and self.bos_id >= 0:
            ids.append(self.bos_id)
        if self.merge_ranks:
            b2u = _bytes_to_unicode()
            mapped = "".join(b2u[b] for b in text.encode("utf-8"))
            for piece in self._bpe(mapped):
                tid = self.vocab.get(piece)
                if tid is not None:
                    ids.append(tid)
                else:
                    for ch in piece:
                        tid = self.vocab.get(ch)
                        if tid is not None:
                            ids.append(tid)
        elif self.byte_tokens:
            for b in text.encode("utf-8"):
                tid = self.byte_tokens.get(b)
                if tid is not None:
                    ids.append(tid)
        else:
            for ch in text:
                tid = self.vocab.get(ch)
                if tid is not None:
                    ids.append(tid)
        return ids

    # ------------------------------------------------------------- decode

    def decode(self, ids: list[int]) -> str:
        out = bytearray()
        b2u = _bytes_to_unicode() if self.merge_ranks else None
        u2b = {v: k for k, v in b2u.items()} if b2u else None
        rev_byte = {v: k for k, v in self.byte_tokens.items()}
        for i in ids:
            if i in (self.bos_id, self.eos_id):
                continue
            if not (0 <= i < len(self.tokens)):
                continue
            tok = self.tokens[i]
            if i in rev_byte:
                out.append(rev_byte[i])
            elif u2b is not None:
                for ch in tok:
                    if ch in u2b:
                        out.append(u2b[ch])
                    else:
                        out.extend(ch.encode("utf-8"))
            else:
                out.extend(tok.replace("▁", " ").encode("utf-8"))
        return out.decode("utf-8", errors="replace")

    def __len__(self) -> int:
        return len(self.tokens)


class NativeTokenizer:
    """C++ tokenizer from the engine extension (ops/csrc/tokenizer.cpp)."""

    def __init__(self, tokens, merges=None, bos_id=1, eos_id=2):
        from .ops import get_core
        core = get_core()
        self._t = core.Tokenizer(list(tokens), list(merges or []),
                                 bos_id, eos_id)
        self.bos_id = bos_id
        self.eos_id = eos_id

    @classmethod
    def from_gguf(cls, reader) -> "NativeTokenizer":
        md = reader.metadata
        return cls(tokens=md.get("tokenizer.ggml.tokens", []),
                   merges=md.get("tokenizer.ggml.merges"),
                   bos_id=md.get("tokenizer.ggml.bos_token_id", 1),
                   eos_id=md.get("tokenizer.ggml.eos_token_id", 2))

    def encode(self, text: str, add_bos: bool = True) -> list[int]:
        return list(self._t.encode(text, add_bos))

    def decode(self, ids: list[int]) -> str:
        return self._t.decode(list(ids)).decode("utf-8", errors="replace")

    def __len__(self) -> int:
        return len(self._t)
