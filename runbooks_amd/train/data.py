"""Training data: jsonl instruction datasets (the reference's dataset
artifact format — SURVEY.md §2b "dataset-loader image") and synthetic
token streams for benchmarking (no network in bench environments)."""
from __future__ import annotations

import json
from pathlib import Path

import torch


class SyntheticTokens(torch.utils.data.Dataset):
    """Random token sequences of fixed shape — bench.py's data source."""

    def __init__(self, vocab_size: int, seq_len: int, n: int = 1 << 16, seed: int = 0):
        self.vocab = vocab_size
        self.seq = seq_len
        self.n = n
        self.seed = seed

    def __len__(self):
        return self.n

    def __getitem__(self, i):
        g = torch.Generator().manual_seed(self.seed * 1_000_003 + i)
        return torch.randint(0, self.vocab, (self.seq,), generator=g)


class JsonlTextDataset(torch.utils.data.Dataset):
    """Reads {"text": ...} or {"prompt","completion"} jsonl, tokenizes with a
    byte-level fallback tokenizer when none is supplied."""

    def __init__(self, path: str | Path, seq_len: int, tokenizer=None,
                 vocab_size: int | None = None):
        self.rows = []
        with open(path) as f:
            for line in f:
                line = line.strip()
                if line:
                    self.rows.append(json.loads(line))
        self.seq = seq_len
        self.tok = tokenizer
        # clamp ids into the model's vocab (a byte-fallback tokenizer can
        # emit ids past a tiny model's embedding table)
        self.vocab = vocab_size

    def __len__(self):
        return len(self.rows)

    def _encode(self, text: str) -> list[int]:
        if self.tok is not None:
            return self.tok.encode(text)
        return list(text.encode("utf-8"))

    def __getitem__(self, i):
        row = self.rows[i]
        text = row.get("text") or (row.get("prompt", "") + row.get("completion", ""))
        ids = self._encode(text)[: self.seq]
        if self.vocab is not None:
            ids = [i % self.vocab for i in ids]
        ids = ids + [0] * (self.seq - len(ids))
        return torch.tensor(ids, dtype=torch.long)


def data_loader(dataset, batch_size: int, rank: int = 0, world: int = 1,
                seed: int = 0, drop_last: bool = True):
    sampler = torch.utils.data.distributed.DistributedSampler(
        dataset, num_replicas=world, rank=rank, shuffle=True, seed=seed,
        drop_last=drop_last) if world > 1 else None
    return torch.utils.data.DataLoader(
        dataset, batch_size=batch_size, sampler=sampler,
        shuffle=(sampler is None), drop_last=drop_last)
