"""Tokenizer loading with a byte-level fallback.

Real models mounted at /content/model carry their tokenizer files
(tokenizer.json / tokenizer.model); synthetic random-weight serving uses
the byte fallback so the whole stack runs with no downloads.
"""
from __future__ import annotations

from pathlib import Path


class ByteTokenizer:
    """Reversible byte-level tokenizer (vocab 256 + bos)."""

    vocab_size = 257
    bos_id = 256
    eos_id = None

    def encode(self, text: str) -> list[int]:
        return [self.bos_id] + list(text.encode("utf-8"))

    def decode(self, ids) -> str:
        return bytes(i for i in ids if 0 <= i < 256).decode("utf-8", errors="replace")


class HFTokenizer:
    def __init__(self, path: str):
        from tokenizers import Tokenizer
        self.tok = Tokenizer.from_file(path)
        # common eos conventions across llama/mistral/qwen/falcon vocabs
        self.eos_id = next(
            (i for i in (self.tok.token_to_id(t) for t in
                         ("</s>", "<|endoftext|>", "<|end_of_text|>",
                          "<|im_end|>", "<eos>"))
             if i is not None), None)

    @property
    def vocab_size(self):
        return self.tok.get_vocab_size()

    def encode(self, text: str) -> list[int]:
        return self.tok.encode(text).ids

    def decode(self, ids) -> str:
        return self.tok.decode(list(ids))


class SPTokenizer:
    def __init__(self, path: str):
        import sentencepiece as spm
        self.sp = spm.SentencePieceProcessor(model_file=path)
        self.eos_id = self.sp.eos_id() if self.sp.eos_id() >= 0 else None

    @property
    def vocab_size(self):
        return self.sp.vocab_size()

    def encode(self, text: str) -> list[int]:
        return self.sp.encode(text)

    def decode(self, ids) -> str:
        return self.sp.decode(list(ids))


def load_tokenizer(model_dir: str | Path | None):
    if model_dir is not None:
        d = Path(model_dir)
        if (d / "tokenizer.json").exists():
            return HFTokenizer(str(d / "tokenizer.json"))
        if (d / "tokenizer.model").exists():
            return SPTokenizer(str(d / "tokenizer.model"))
    return ByteTokenizer()
