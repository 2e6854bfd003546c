"""Tokenizer wrappers with one uniform interface.

Reference parity: python/hetu/data/tokenizers/ (GPT2 BPE, SentencePiece,
tiktoken, HF wrappers).  Offline-friendly: HF/SentencePiece load from local
files; ByteTokenizer needs no assets at all (useful for tests/synthetic
corpora).
"""
from __future__ import annotations

from typing import List


class Tokenizer:
    vocab_size: int

    def encode(self, text: str) -> List[int]:
        raise NotImplementedError

    def decode(self, ids: List[int]) -> str:
        raise NotImplementedError


class ByteTokenizer(Tokenizer):
    """UTF-8 bytes + <bos>/<eos>/<pad> specials; vocab 259."""

    PAD, BOS, EOS = 256, 257, 258

    def __init__(self):
        self.vocab_size = 259

    def encode(self, text: str, bos: bool = False, eos: bool = False
               ) -> List[int]:
        ids = list(text.encode("utf-8"))
        if bos:
            ids = [self.BOS] + ids
        if eos:
            ids = ids + [self.EOS]
        return ids

    def decode(self, ids: List[int]) -> str:
        return bytes(i for i in ids if i < 256).decode("utf-8",
                                                       errors="replace")


class HFTokenizer(Tokenizer):
    """Any local HuggingFace tokenizer dir/file (tokenizers package)."""

    def __init__(self, path: str):
        from transformers import AutoTokenizer
        self.tok = AutoTokenizer.from_pretrained(path)
        self.vocab_size = len(self.tok)

    def encode(self, text: str) -> List[int]:
        return self.tok.encode(text)

    def decode(self, ids: List[int]) -> str:
        return self.tok.decode(ids)


class SentencePieceTokenizer(Tokenizer):
    def __init__(self, model_path: str):
        import sentencepiece as spm
        self.sp = spm.SentencePieceProcessor(model_file=model_path)
        self.vocab_size = self.sp.vocab_size()

    def encode(self, text: str) -> List[int]:
        return self.sp.encode(text)

    def decode(self, ids: List[int]) -> str:
        return self.sp.decode(ids)


def build_tokenizer(kind: str, path: str = "") -> Tokenizer:
    if kind == "byte":
        return ByteTokenizer()
    if kind == "hf":
        return HFTokenizer(path)
    if kind == "sentencepiece":
        return SentencePieceTokenizer(path)
    raise ValueError(f"unknown tokenizer kind {kind}")
