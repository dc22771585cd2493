"""Serving: paged-KV engine, HTTP completions API (container contract)."""
from .engine import BlockAllocator, Engine, Request  # noqa: F401
from .tokenizer import ByteTokenizer, load_tokenizer  # noqa: F401
