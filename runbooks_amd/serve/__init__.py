"""Serving: paged-KV engine, HTTP completions API (container contract).

``build_app`` / ``serve_forever`` import lazily (fastapi optional for
pure-engine users).
"""
from .engine import BlockAllocator, Engine, Request  # noqa: F401
from .tokenizer import ByteTokenizer, load_tokenizer  # noqa: F401


def build_app(*args, **kwargs):
    from .http import build_app as _b
    return _b(*args, **kwargs)


def serve_forever(*args, **kwargs):
    from .http import serve_forever as _s
    return _s(*args, **kwargs)
