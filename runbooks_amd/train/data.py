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


TEXT_SUFFIXES = (".jsonl", ".parquet", ".csv", ".txt")


def _load_rows(path: Path) -> list[dict]:
    suf = path.suffix.lower()
    if suf == ".jsonl":
        rows = []
        with open(path) as f:
            for line in f:
                line = line.strip()
                if line:
                    rows.append(json.loads(line))
        return rows
    if suf == ".parquet":
        import pyarrow.parquet as pq
        return pq.read_table(path).to_pylist()
    if suf == ".csv":
        import csv
        with open(path, newline="") as f:
            return list(csv.DictReader(f))
    if suf == ".txt":
        with open(path) as f:
            return [{"text": ln.rstrip("\n")} for ln in f if ln.strip()]
    raise ValueError(f"unsupported dataset file {path} "
                     f"(supported: {TEXT_SUFFIXES})")


class TextDataset(torch.utils.data.Dataset):
    """Instruction/text rows from jsonl / parquet / csv / txt files —
    the formats dataset-loader jobs commonly produce ({"text": ...} or
    {"prompt","completion"} rows). Accepts one file or a directory (all
    supported files, sorted). Tokenizes with a byte-level fallback when
    no tokenizer is supplied."""

    def __init__(self, path: str | Path, seq_len: int, tokenizer=None,
                 vocab_size: int | None = None):
        p = Path(path)
        files = (sorted(f for f in p.iterdir()
                        if f.suffix.lower() in TEXT_SUFFIXES)
                 if p.is_dir() else [p])
        if not files:
            raise FileNotFoundError(f"no dataset files under {p}")
        self.rows = [row for f in files for row in _load_rows(f)]
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


# legacy name (the class now reads all supported formats)
JsonlTextDataset = TextDataset


def data_loader(dataset, batch_size: int, rank: int = 0, world: int = 1,
                seed: int = 0, drop_last: bool = True):
    sampler = torch.utils.data.distributed.DistributedSampler(
        dataset, num_replicas=world, rank=rank, shuffle=True, seed=seed,
        drop_last=drop_last) if world > 1 else None
    return torch.utils.data.DataLoader(
        dataset, batch_size=batch_size, sampler=sampler,
        shuffle=(sampler is None), drop_last=drop_last)
