"""Tokenizer tests: Python reference + native C++ parity."""

import pytest

from crowdllama_amd.tokenizer import Tokenizer


BYTE_TOKENS = ["<unk>", "<s>", "</s>"] + [f"<0x{i:02X}>" for i in range(256)]


def test_byte_fallback_roundtrip():
    t = Tokenizer(BYTE_TOKENS)
    for text in ["hello", "héllo wörld", "你好", "a b\nc"]:
        ids = t.encode(text)
        assert ids[0] == t.bos_id
        assert t.decode(ids) == text


def test_bpe_merges():
    tokens = BYTE_TOKENS + ["he", "ll", "hell", "hello"]
    merges = ["h e", "l l", "he ll", "hell o"]
    t = Tokenizer(tokens, merges=merges)
    ids = t.encode("hello", add_bos=False)
    assert ids == [t.vocab["hello"]]
    assert t.decode(ids) == "hello"


def test_native_matches_python():
    pytest.importorskip("crowdllama_amd.ops._core")
    from crowdllama_amd.tokenizer import NativeTokenizer, _bytes_to_unicode
    # BPE vocab with full single-char coverage + a few merges
    chars = sorted(set(_bytes_to_unicode().values()))
    tokens = ["<unk>", "<s>", "</s>"] + chars + ["he", "ll", "hell"]
    merges = ["h e", "l l", "he ll"]
    py = Tokenizer(tokens, merges=merges)
    nat = NativeTokenizer(tokens, merges=merges)
    for text in ["hello world", "h\u00e9llo", "abc!", "hell"]:
        assert nat.encode(text) == py.encode(text), text
        assert nat.decode(nat.encode(text)) == text
        assert py.decode(py.encode(text)) == text


def test_native_byte_fallback():
    pytest.importorskip("crowdllama_amd.ops._core")
    from crowdllama_amd.tokenizer import NativeTokenizer
    nat = NativeTokenizer(BYTE_TOKENS)
    py = Tokenizer(BYTE_TOKENS)
    for text in ["hello", "h\u00e9llo w\u00f6rld", "\u4f60\u597d"]:
        assert nat.encode(text) == py.encode(text)
        assert nat.decode(nat.encode(text)) == text
